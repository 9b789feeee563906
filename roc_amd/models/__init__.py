from .gcn import GCN
from .sage import GraphSAGE
from .gin import GIN
from .gat import GAT
from .sgc import SGC
from .appnp import APPNP


def build_model(name: str, dims, dropout: float = 0.5, seed: int = 1, **kw):
    name = name.lower()
    if name == "gcn":
        return GCN(dims, dropout=dropout, seed=seed, **kw)
    if name in ("sage", "graphsage"):
        return GraphSAGE(dims, dropout=dropout, seed=seed, **kw)
    if name == "gin":
        return GIN(dims, dropout=dropout, seed=seed, **kw)
    if name == "gat":
        return GAT(dims, dropout=dropout, seed=seed, **kw)
    if name == "sgc":
        return SGC(dims, dropout=dropout, seed=seed, **kw)
    if name == "appnp":
        return APPNP(dims, dropout=dropout, seed=seed, **kw)
    raise ValueError(f"unknown model {name!r}")
