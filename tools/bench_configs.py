#!/usr/bin/env python3
"""Benchmark the non-headline BASELINE.json configs (bench.py covers the
headline Wide&Deep config 3):

  --config 2   : 5-layer MLP, 200 dense numerics, bf16, 1 GPU
  --config 4   : DeepFM (26x1M-vocab embeddings + 200 dense), 1 GPU
  --config 5   : 1B-param deep tower (200 dense -> 8192 x 14 + head), 1 GPU
  --config 5io : 100M-row-class ingest rate of the native CSV reader (CPU ok;
                 rows scaled by --rows, rate extrapolates)

Each prints one JSON line.  Usage: python tools/bench_configs.py --config 2
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch


def _train_throughput(model, batches, steps, warmup, optimizer="adam", lr=1e-3):
    from shifu_amd.ops.flat import FlatParams, split_params
    from shifu_amd.ops.loss import weighted_loss
    from shifu_amd.ops.optim import FusedOptimizer
    dense_params, emb_params = split_params(model)
    from shifu_amd.ops.flat import bind_mirrors
    flat = FlatParams(dense_params, mirror_bf16=torch.cuda.is_available())
    bind_mirrors(model, flat)
    opt = FusedOptimizer(flat, emb_params, optimizer=optimizer, lr=lr)

    def eager_step(i):
        dense, cats, target, weight = batches[i % len(batches)]
        loss = weighted_loss(model(dense, cats), target, weight, "sigmoid_ce")
        loss.backward()
        flat.sync_grads()
        opt.step()
        opt.zero_grad()
        return loss

    step = eager_step
    if torch.cuda.is_available():
        # hipGraph the step (same machinery as bench.py): the eager loop is
        # host-launch-bound on these sub-ms models and measured 12-14M
        # samples/s box-to-box noise on identical code; graphed runs are stable.
        try:
            sd, sc, st, sw = [x.clone() for x in batches[0]]

            def body():
                loss = weighted_loss(model(sd, sc), st, sw, "sigmoid_ce")
                loss.backward()
                flat.sync_grads()
                opt.step()
                opt.zero_grad()

            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    body()
            torch.cuda.current_stream().wait_stream(side)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                body()

            def graphed_step(i):
                dense, cats, target, weight = batches[i % len(batches)]
                sd.copy_(dense)
                st.copy_(target)
                sw.copy_(weight)
                if sc.numel():
                    sc.copy_(cats)
                graph.replay()

            step = graphed_step
        except Exception as e:  # capture unsupported -> eager numbers
            print(f"# hipGraph capture failed ({e}); falling back to eager",
                  flush=True)

    for i in range(warmup):
        step(i)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.time()
    for i in range(steps):
        step(warmup + i)
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return time.time() - t0


def _batches(B, n_dense, vocab, device, dtype, n=4, seed=0):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n):
        dense = torch.randn(B, n_dense, generator=g).to(device=device, dtype=dtype)
        cats = (torch.randint(0, max(vocab) if vocab else 1, (B, len(vocab)),
                              generator=g).to(device)
                if vocab else torch.zeros(B, 0, dtype=torch.int64, device=device))
        target = (torch.rand(B, generator=g) > 0.5).float().to(device)
        weight = torch.ones(B, device=device)
        out.append((dense, cats, target, weight))
    return out


def _emit(name, samples, elapsed, steps, extra):
    print(json.dumps({
        "metric": f"samples_per_sec_{name}", "value": samples / elapsed,
        "unit": "samples/s", "steps": steps,
        "ms_per_step": elapsed / steps * 1000.0,
        "higher_is_better": True, "data": "synthetic", **extra}), flush=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", required=True, choices=["1", "2", "4", "5", "5io"])
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=8192)
    ap.add_argument("--rows", type=int, default=2_000_000)
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    device = "cuda" if on_gpu else "cpu"
    dtype = torch.bfloat16 if on_gpu else torch.float32
    torch.manual_seed(1)

    if args.config == "1":
        # config 1: 3-layer MLP on 1k-row synthetic CSV, CPU world_size=1 —
        # exercises the FULL plumbing (CSV parse, ModelConfig, Trainer,
        # checkpoint, export) and reports end-to-end wall time.
        import tempfile
        from shifu_amd.config.model_config import ModelConfig
        from shifu_amd.config.run_config import RunConfig
        from shifu_amd.data.synthetic import generate_synthetic_csv
        from shifu_amd.io import load_csv_native
        from shifu_amd.models.mlp import ShifuMLP
        from shifu_amd.train.trainer import Trainer
        mc = ModelConfig.from_dict({"train": {"numTrainEpochs": 3, "params": {
            "NumHiddenLayers": 3, "NumHiddenNodes": [64, 32, 16],
            "ActivationFunc": ["relu"] * 3, "LearningRate": 0.01,
            "Optimizer": "adam", "Loss": "sigmoid_ce", "MiniBatchSize": 100,
            "L2Reg": 0.0}}})
        with tempfile.TemporaryDirectory() as td:
            paths = generate_synthetic_csv(td + "/data", n_rows=1000,
                                           n_dense=30, n_files=1, seed=1)
            t0 = time.time()
            ds = load_csv_native(paths, list(range(2, 32)), [], 0, 1)
            train, valid = ds.split(0.2, seed=1)
            rc = RunConfig(tmp_model_path=td + "/ckpt",
                           final_model_path=td + "/final", device="cpu")
            tr = Trainer(ShifuMLP(30, [64, 32, 16], ["relu"] * 3, seed=1),
                         mc, rc, train, valid)
            tr.fit()
            el = time.time() - t0
            auc = tr.evaluate(tr.valid_data)["auc"]
        print(json.dumps({
            "metric": "config1_cpu_e2e_seconds", "value": el, "unit": "s",
            "higher_is_better": False, "data": "synthetic",
            "config": {"model": "mlp[30->64,32,16]", "rows": 1000,
                       "epochs": 3, "auc": auc}}), flush=True)
        return

    if args.config == "2":
        from shifu_amd.models.mlp import ShifuMLP
        model = ShifuMLP(200, [1024, 512, 256, 128, 64],
                         ["relu"] * 5, seed=2).to(device)
        b = _batches(args.batch, 200, [], device, dtype)
        el = _train_throughput(model, b, args.steps, args.warmup)
        _emit("mlp5_200d", args.batch * args.steps, el, args.steps,
              {"config": {"model": "mlp[200->1024,512,256,128,64]",
                          "batch": args.batch, "dtype": str(dtype)}})

    elif args.config == "4":
        from shifu_amd.models.deepfm import DeepFM
        from shifu_amd.ops.embedding import UnifiedMultiEmbedding
        vocab = [1_000_000] * 26
        model = DeepFM(200, vocab, 64, [1024, 512, 256], ["relu"] * 3,
                       seed=3, unified=True).to(device)
        if on_gpu:
            for p in model.parameters():
                if getattr(p, "_is_embedding_arena", False):
                    p.data = p.data.to(torch.bfloat16)
        for m in model.modules():
            if isinstance(m, UnifiedMultiEmbedding):
                m.defer_grads = True   # single-rank unpacked-grad fast path
        b = _batches(args.batch, 200, vocab, device, dtype)
        el = _train_throughput(model, b, args.steps, args.warmup)
        _emit("deepfm", args.batch * args.steps, el, args.steps,
              {"config": {"model": "deepfm[26x1Mvocab*64d+200dense]",
                          "batch": args.batch, "dtype": str(dtype)}})

    elif args.config == "5":
        from shifu_amd.models.mlp import ShifuMLP
        hidden = [8192] * 15 + [1024]
        model = ShifuMLP(200, hidden, ["relu"] * len(hidden), seed=4).to(device)
        n_params = sum(p.numel() for p in model.parameters())
        B = min(args.batch, 4096)
        b = _batches(B, 200, [], device, dtype, n=2)
        el = _train_throughput(model, b, max(args.steps // 2, 3),
                               max(args.warmup // 2, 1))
        _emit("tower1b", B * max(args.steps // 2, 3), el, max(args.steps // 2, 3),
              {"config": {"model": f"mlp[{n_params/1e9:.2f}B params, 16 hidden]",
                          "batch": B, "dtype": str(dtype)}})

    elif args.config == "5io":
        import tempfile
        from shifu_amd.data.synthetic import generate_synthetic_csv
        from shifu_amd.io import load_csv_native, native_io
        with tempfile.TemporaryDirectory() as td:
            paths = generate_synthetic_csv(td, n_rows=args.rows, n_dense=200,
                                           n_files=32, seed=7)
            t0 = time.time()
            ds = load_csv_native(paths, selected_numeric=list(range(2, 202)),
                                 target_column=0, weight_column=1)
            el = time.time() - t0
        print(json.dumps({
            "metric": "csv_ingest_rows_per_sec", "value": len(ds) / el,
            "unit": "rows/s", "rows": len(ds), "elapsed_s": el,
            "native": native_io() is not None,
            "extrapolated_100M_minutes": 100e6 / (len(ds) / el) / 60.0,
        }), flush=True)


if __name__ == "__main__":
    main()
