#!/usr/bin/env python3
"""Training driver — the framework's CLI (reference `gnn.cc:25-179`).

Single process:
  python train.py --dataset cora-synthetic --layers 1433-16-7 --epochs 200
Multi GPU (one process per GPU, RCCL):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 train.py --dataset reddit-synthetic ...

Reference flag parity: --file (.lux dataset prefix), --layers D0-D1-...-C,
--epochs, --lr, --weight-decay, --decay-rate, --decay-steps, --dropout,
--seed; metrics printed every 5 epochs (`gnn.cc:107-110`). New:
--model gcn|sage|gin|gat, --dtype, --checkpoint/--resume, --trace.
"""
import argparse
import os
import sys
import time

import torch
import torch.distributed as dist

from roc_amd import (synthetic_dataset, build_shard, build_model,
                     AdamOptimizer, Trainer)
from roc_amd.graph import (load_lux, load_features, load_labels, load_mask,
                           DATASET_SHAPES)
from roc_amd.parallel.partition import edge_balanced_bounds
from roc_amd.utils import save_checkpoint, load_checkpoint
from roc_amd.debug import check_metrics, TrainingDiverged


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default=None,
                    help="YAML config (CLI flags override; see configs/)")
    ap.add_argument("--file", default=None,
                    help=".lux dataset prefix (expects <p>.add_self_edge.lux,"
                         " <p>.feats.csv/.bin, <p>.label, <p>.mask)")
    ap.add_argument("--dataset", default="cora-synthetic",
                    help="<name>-synthetic for generated data; see "
                         "roc_amd.graph.DATASET_SHAPES")
    ap.add_argument("--scale", type=float, default=1.0)
    ap.add_argument("--layers", default=None,
                    help="dash-separated dims incl. input and classes, "
                         "e.g. 602-256-41")
    ap.add_argument("--hidden", type=int, default=256)
    ap.add_argument("--num-layers", type=int, default=2)
    ap.add_argument("--model", default="gcn", choices=["gcn", "sage", "gin", "gat", "sgc", "appnp"])
    ap.add_argument("--epochs", "-e", type=int, default=100)
    ap.add_argument("--lr", type=float, default=0.01)
    ap.add_argument("--weight-decay", "--wd", type=float, default=1e-4)
    ap.add_argument("--decay-rate", type=float, default=0.97)
    ap.add_argument("--decay-steps", type=int, default=100)
    ap.add_argument("--dropout", type=float, default=0.5)
    ap.add_argument("--seed", type=int, default=1)
    ap.add_argument("--dtype", default="auto", choices=["auto", "bf16", "fp32"])
    ap.add_argument("--loss", default="mean", choices=["mean", "sum"],
                    help="sum replicates the reference's unscaled gradient")
    ap.add_argument("--eval-every", type=int, default=5)
    ap.add_argument("--residual", action="store_true")
    ap.add_argument("--heads", type=int, default=4,
                    help="GAT attention heads (concat; last layer 1)")
    ap.add_argument("--k-hops", type=int, default=None,
                    help="SGC propagation depth (default 2) / APPNP "
                         "propagation steps (default 10)")
    ap.add_argument("--alpha", type=float, default=0.1,
                    help="APPNP teleport probability")
    ap.add_argument("--checkpoint", default=None)
    ap.add_argument("--checkpoint-every", type=int, default=0)
    ap.add_argument("--resume", default=None)
    ap.add_argument("--trace", default=None,
                    help="write a chrome trace JSON here at the end")
    ap.add_argument("--save-best", default=None,
                    help="track the best val_acc at each eval and keep a "
                         "checkpoint of it here")
    ap.add_argument("--export-safetensors", default=None,
                    help="after training, write the model weights (fp32 "
                         "masters) as a .safetensors file")
    ap.add_argument("--predict", default=None,
                    help="inference mode: skip training, run one forward "
                         "in infer mode (dropout=identity) and write "
                         "per-node predicted class ids (.npy, int64; "
                         "multi-rank: one <path>.rankN.npy per rank with "
                         "<path>.bounds.npy). Pair with --resume.")
    ap.add_argument("--offload", action="store_true",
                    help="host-DRAM activation offload (capacity tier)")
    ap.add_argument("--recompute", action="store_true",
                    help="per-layer activation recompute (capacity tier: "
                         "O(1-layer) transient HBM, no pinned host memory; "
                         "exact — dropout masks regenerate identically)")
    ap.add_argument("--learnable-labels", action="store_true",
                    help="synthetic labels from a random one-hop teacher "
                         "(accuracy can actually rise)")
    ap.add_argument("--locality", type=float, default=0.0,
                    help="synthetic graphs: fraction of edges kept inside "
                         "a planted community (community structure like "
                         "real graphs; see --reorder)")
    ap.add_argument("--communities", type=int, default=64,
                    help="synthetic graphs: number of planted communities")
    ap.add_argument("--reorder", default="none",
                    choices=["none", "degree", "rcm", "cluster", "auto"],
                    help="locality relabeling applied to the whole dataset "
                         "before partitioning (sparser halos, denser SpMM "
                         "gathers); not available with windowed --file "
                         "loading (world>1 + --file)")
    ap.add_argument("--rebalance-every", type=int, default=0,
                    help="cost-model repartition every N epochs (measured "
                         "per-rank throughput; multi-rank only)")
    ap.add_argument("--sample", default=None,
                    help="mini-batch neighbor sampling: per-layer fanouts "
                         "e.g. '25,10' (layer order, input-side first); "
                         "GraphSAGE only, world_size=1")
    ap.add_argument("--batch-size", type=int, default=1024,
                    help="target nodes per sampled mini-batch")
    ap.add_argument("--log-csv", default=None,
                    help="append one CSV row per eval (epoch, losses, "
                         "accuracies, ms/epoch); rank 0 only")
    ap.add_argument("--clip-norm", type=float, default=0.0,
                    help="global gradient-norm clip (0 = off); applied to "
                         "the all-reduced flat gradient on every rank")
    ap.add_argument("--auto-recover", action="store_true",
                    help="on divergence (non-finite loss): restore the "
                         "last checkpoint, halve the LR, continue")
    ap.add_argument("-v", "--verbose", action="store_true")
    args = ap.parse_args()
    if args.config:
        import yaml
        with open(args.config) as f:
            cfg = yaml.safe_load(f) or {}
        defaults = {}
        for k, v in cfg.items():
            key = k.replace("-", "_")
            assert hasattr(args, key), f"unknown config key {k}"
            defaults[key] = v
        # CLI flags that were explicitly set keep priority
        import sys as _sys
        given = {a.split("=")[0].lstrip("-").replace("-", "_")
                 for a in _sys.argv[1:] if a.startswith("--")}
        for k, v in defaults.items():
            if k not in given:
                setattr(args, k, v)
    return args


def load_dataset(args, rank=0, world=1):
    """Synthetic shapes, or reference-format files. With --file and
    world > 1, each rank reads ONLY its window of the edge list and
    feature matrix (reference `load_task.cu:231-243` parity); the halo
    send plan is exchanged over the process group."""
    if args.file:
        assert args.layers, "--file requires --layers D0-...-C"
        dims = [int(d) for d in args.layers.split("-")]
        lux = args.file + ".add_self_edge.lux"
        if world > 1:
            from roc_amd.parallel.partition import build_shard_from_lux
            from roc_amd.graph import load_features_window, load_lux_meta
            shard = build_shard_from_lux(lux, rank, world)
            n = load_lux_meta(lux)[0]
            feats = load_features_window(args.file, n, dims[0],
                                         shard.lo, shard.hi)
            labels = load_labels(args.file + ".label", n)[shard.lo:shard.hi]
            mask = load_mask(args.file + ".mask", n)[shard.lo:shard.hi]
            return (None, feats, labels, mask, dims[-1]), shard
        g = load_lux(lux)
        feats = load_features(args.file, g.num_nodes, dims[0])
        labels = load_labels(args.file + ".label", g.num_nodes)
        mask = load_mask(args.file + ".mask", g.num_nodes)
        return (g, feats, labels, mask, dims[-1]), None
    name = args.dataset.replace("-synthetic", "")
    assert name in DATASET_SHAPES, f"unknown dataset {name}"
    return synthetic_dataset(name, seed=args.seed, scale=args.scale,
                             learnable_labels=args.learnable_labels,
                             locality=args.locality,
                             num_communities=args.communities), None


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    on_gpu = torch.cuda.is_available()
    if world > 1:
        local_rank = int(os.environ.get("LOCAL_RANK", rank))
        if on_gpu:
            # pin BEFORE init so RCCL binds the right device
            torch.cuda.set_device(local_rank)
        device = f"cuda:{local_rank}" if on_gpu else "cpu"
        dist.init_process_group(
            "nccl" if on_gpu else "gloo", rank=rank, world_size=world,
            device_id=torch.device(device) if on_gpu else None)
    else:
        device = "cuda:0" if on_gpu else "cpu"

    (g, feats, labels, mask, num_classes), pre_shard = \
        load_dataset(args, rank, world)
    if args.reorder != "none":
        if g is None:
            sys.exit("--reorder needs the whole graph in memory; it is not "
                     "available with windowed --file loading (world>1)")
        from roc_amd.graph import apply_ordering
        g, feats, labels, mask, _ = apply_ordering(
            g, feats, labels, mask, args.reorder)
    # pad feature dim for 16-B-aligned bf16 rows (zero cols; exact math)
    pad = (-feats.shape[1]) % 8
    if pad and on_gpu:
        feats = torch.nn.functional.pad(feats, (0, pad))

    if args.layers:
        dims = [int(d) for d in args.layers.split("-")]
        dims[0] = feats.shape[1]
        if len(dims) < 2 or dims[-1] < num_classes:
            sys.exit(f"--layers must give >=2 dash-separated dims ending "
                     f">= num_classes={num_classes}, e.g. 602-256-41 "
                     f"(got {args.layers!r}; for a layer COUNT use "
                     f"--num-layers N)")
    else:
        c_out = num_classes
        if on_gpu:  # pad class dim to whole 128-B logits rows (see bench.py)
            c_out = num_classes + ((-num_classes) % 64)
        dims = [feats.shape[1]] + [args.hidden] * (args.num_layers - 1) \
            + [c_out]

    if pre_shard is not None:
        shard = pre_shard
    else:
        bounds = edge_balanced_bounds(g.rowptr, world)
        shard = build_shard(g, rank, world, bounds,
                            use_comm=(world > 1 and dist.is_initialized()))
    mkw = {"residual": args.residual} if args.model == "gcn" else {}
    if args.model == "gat":
        mkw = {"heads": args.heads}
    elif args.model == "sgc" and args.k_hops:
        mkw = {"k": args.k_hops}
    elif args.model == "appnp":
        mkw = {"alpha": args.alpha}
        if args.k_hops:
            mkw["k"] = args.k_hops
    model = build_model(args.model, dims, dropout=args.dropout,
                        seed=args.seed, **mkw)
    if args.recompute:
        assert not args.offload, "--recompute and --offload are exclusive " \
            "(both re-route activation residency)"
        model.recompute = True
    opt = AdamOptimizer(model.parameters(), lr=args.lr,
                        weight_decay=args.weight_decay,
                        decay_rate=args.decay_rate,
                        decay_steps=args.decay_steps)
    dtype = torch.float32
    if args.dtype == "bf16" or (args.dtype == "auto" and on_gpu):
        dtype = torch.bfloat16
    n_train = int((mask == 1).sum())
    if world > 1 and dist.is_initialized() and pre_shard is not None:
        # windowed loading: mask is the LOCAL slice, so sum the train
        # counts so every rank uses the GLOBAL 1/n_train grad scale
        # (otherwise the summed all-reduced gradient is rank-weighted)
        from roc_amd.parallel.comm import allreduce_scalar_int
        n_train = allreduce_scalar_int(n_train)
    gs = 1.0 if args.loss == "sum" else 1.0 / max(n_train, 1)
    if args.sample:
        # sampled mini-batch tier (roc_amd/sampling.py): host-resident
        # graph+features, bounded device memory per step
        from roc_amd.sampling import MiniBatchTrainer
        assert world == 1, "--sample is a single-process mode"
        assert args.model == "sage", "--sample needs --model sage"
        assert g is not None, "--sample needs a full-graph dataset"
        assert not (args.predict or args.save_best or args.log_csv or
                    args.resume), (
            "--predict/--save-best/--log-csv/--resume are full-graph-loop "
            "features: train with --sample --checkpoint, then run them in "
            "a separate invocation without --sample")
        fanouts = [int(f) for f in args.sample.split(",")]
        assert len(fanouts) == len(dims) - 1, \
            f"need one fanout per layer ({len(dims) - 1})"
        mb = MiniBatchTrainer(model, g, feats, labels, mask, opt,
                              fanouts=fanouts, batch_size=args.batch_size,
                              device=device, compute_dtype=dtype,
                              seed=args.seed, num_classes=num_classes
                              if dims[-1] != num_classes else None)
        import time as _time
        t0 = _time.perf_counter()
        for ep in range(1, args.epochs + 1):
            loss = mb.train_epoch()
            if args.eval_every and ep % args.eval_every == 0:
                md = mb.evaluate(shard)
                dt = _time.perf_counter() - t0
                print(f"epoch {ep:5d}  batch-loss {loss:.4f}  "
                      f"train {md['train_acc']:.4f}  "
                      f"val {md['val_acc']:.4f}  "
                      f"[{dt / ep * 1e3:.1f} ms/epoch]", flush=True)
        if args.checkpoint:
            trainer = Trainer(model, shard, feats, labels, mask, opt,
                              device=device, compute_dtype=dtype,
                              seed=args.seed)
            trainer.epoch = args.epochs
            save_checkpoint(args.checkpoint, trainer)
        return

    trainer = Trainer(model, shard, feats, labels, mask, opt, device=device,
                      compute_dtype=dtype, grad_scale=gs, seed=args.seed,
                      num_classes=num_classes if dims[-1] != num_classes
                      else None, local_slices=pre_shard is not None)
    trainer.clip_norm = args.clip_norm
    if args.trace:
        trainer.enable_tracing()
    if args.offload:
        trainer.enable_offload()
    if args.resume:
        load_checkpoint(args.resume, trainer)
        if rank == 0:
            print(f"resumed from {args.resume} at epoch {trainer.epoch}")

    if rank == 0:
        nn_ = g.num_nodes if g is not None else shard.bounds[-1]
        ne_ = (g.num_edges if g is not None
               else f"window:{shard.colidx.numel()}")
        print(f"[config] model={args.model} dims={dims} nodes={nn_} "
              f"edges={ne_} world={world} device={device} "
              f"dtype={dtype} lr={args.lr} wd={args.weight_decay} "
              f"dropout={args.dropout} comm={shard.comm_mode}", flush=True)

    if args.rebalance_every:
        if g is not None:
            trainer.attach_full_graph(g)
        else:
            # windowed loading: rebalance re-reads only the new window
            from roc_amd.graph import (load_lux_meta, load_features_window)
            lux = args.file + ".add_self_edge.lux"
            n = load_lux_meta(lux)[0]
            raw_dim = int(args.layers.split("-")[0])  # on-disk width
            padded_dim = feats.shape[1]
            full_labels = load_labels(args.file + ".label", n)
            full_mask = load_mask(args.file + ".mask", n)

            def _loader(lo, hi, _n=n, _d=raw_dim, _p=padded_dim):
                f = load_features_window(args.file, _n, _d, lo, hi)
                if f.shape[1] < _p:
                    f = torch.nn.functional.pad(f, (0, _p - f.shape[1]))
                return f, full_labels[lo:hi], full_mask[lo:hi]

            trainer.attach_windowed_dataset(lux, _loader,
                                            load_lux_meta(lux)[2])

    if args.predict:
        # inference mode: one forward with dropout=identity
        # (reference infer_task semantics, `dropout_kernel.cu:159-180`)
        import numpy as np
        trainer.infer_mode()
        with torch.no_grad():
            logits = trainer.model(trainer.x, trainer.shard, trainer.group)
            nc = num_classes or logits.shape[1]
            pred = logits[:, :nc].float().argmax(dim=1).cpu().numpy()
        if world > 1:
            np.save(f"{args.predict}.rank{rank}.npy", pred)
            if rank == 0:
                np.save(f"{args.predict}.bounds.npy",
                        np.asarray(shard.bounds))
        else:
            np.save(args.predict, pred)
        md = trainer.evaluate()
        if rank == 0:
            print(f"[predict] wrote {args.predict} "
                  f"({pred.shape[0]} local nodes); val_acc "
                  f"{md['val_acc']:.4f} test_acc {md['test_acc']:.4f}",
                  flush=True)
        if world > 1:
            dist.destroy_process_group()
        return

    t_start = time.perf_counter()
    best_val = -1.0
    while trainer.epoch < args.epochs:
        metrics = trainer.train_epoch()
        ep = trainer.epoch
        if args.auto_recover and (args.eval_every == 0
                                  or ep % max(args.eval_every, 1) == 0):
            # the divergence flag is agreed via MAX all-reduce so every
            # rank takes the restore branch together (a NaN confined to
            # one rank's rows must not desynchronize collective counts)
            err = None
            try:
                check_metrics(metrics)
            except TrainingDiverged as e:
                err = e
            bad = int(err is not None)
            if world > 1 and dist.is_initialized():
                from roc_amd.parallel.comm import allreduce_scalar_int
                bad = allreduce_scalar_int(bad, op="max")
            if bad:
                if not (args.checkpoint and os.path.exists(args.checkpoint)):
                    raise err or TrainingDiverged(
                        "divergence detected on a peer rank")
                if rank == 0:
                    print(f"[recover] {err or 'peer-rank divergence'}; "
                          f"restoring {args.checkpoint}, "
                          f"lr {trainer.optimizer.lr} -> "
                          f"{trainer.optimizer.lr * 0.5}", flush=True)
                load_checkpoint(args.checkpoint, trainer)
                trainer.optimizer.lr *= 0.5
                continue
        if (args.rebalance_every and ep % args.rebalance_every == 0
                and world > 1):
            nb = trainer.measure_and_rebalance(feats, labels, mask)
            if rank == 0 and args.verbose:
                print(f"[rebalance] epoch {ep}: bounds -> {nb}", flush=True)
        if args.eval_every and ep % args.eval_every == 0:
            md = trainer.evaluate()
            if args.save_best and md["val_acc"] > best_val:
                best_val = md["val_acc"]
                save_checkpoint(args.save_best, trainer,
                                extra={"best_val_acc": best_val,
                                       "best_epoch": ep})
                if rank == 0 and args.verbose:
                    print(f"[best] epoch {ep}: val_acc {best_val:.4f} "
                          f"-> {args.save_best}", flush=True)
            if rank == 0:
                dt = time.perf_counter() - t_start
                print(f"epoch {ep:5d}  loss {md['ce_loss']:.4f}  "
                      f"train {md['train_acc']:.4f}  val {md['val_acc']:.4f}"
                      f"  test {md['test_acc']:.4f}  "
                      f"[{dt / ep * 1e3:.1f} ms/epoch]", flush=True)
                if args.log_csv:
                    new = not os.path.exists(args.log_csv)
                    with open(args.log_csv, "a") as fcsv:
                        if new:
                            fcsv.write("epoch,ce_loss,roc_loss,train_acc,"
                                       "val_acc,test_acc,ms_per_epoch\n")
                        fcsv.write(f"{ep},{md['ce_loss']:.6f},"
                                   f"{md.get('roc_loss', 0.0):.6f},"
                                   f"{md['train_acc']:.6f},"
                                   f"{md['val_acc']:.6f},"
                                   f"{md['test_acc']:.6f},"
                                   f"{dt / ep * 1e3:.3f}\n")
        if (args.checkpoint and args.checkpoint_every
                and ep % args.checkpoint_every == 0):
            save_checkpoint(args.checkpoint, trainer)

    if rank == 0 and on_gpu:
        print(f"[mem] peak HBM allocated "
              f"{torch.cuda.max_memory_allocated() / 2**30:.1f} GiB / "
              f"reserved {torch.cuda.max_memory_reserved() / 2**30:.1f} GiB",
              flush=True)
    if args.checkpoint:
        save_checkpoint(args.checkpoint, trainer)
    if args.export_safetensors and rank == 0:
        from safetensors.torch import save_file
        save_file({k: v.detach().float().cpu().contiguous()
                   for k, v in trainer.model.state_dict().items()},
                  args.export_safetensors)
        print(f"[export] wrote {args.export_safetensors}", flush=True)
    if args.trace and trainer.tracer is not None:
        trainer.tracer.dump_chrome(args.trace, rank)
        if rank == 0:
            summary = trainer.tracer.summarize()
            total = sum(summary.values())
            print("[trace] phase totals (ms): " + ", ".join(
                f"{k}={v:.1f}" for k, v in sorted(summary.items())) +
                f"  (sum {total:.1f})")
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
