"""Run the serving façade:  python -m distributedkernelshap_amd.serve

Loads (or generates) the benchmark dataset + fitted model and serves
/explain with dynamic batching (see serve/app.py).
"""
import argparse
import os


def main():
    p = argparse.ArgumentParser(prog="distributedkernelshap_amd.serve")
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8800)
    p.add_argument("--max-batch-size", type=int, default=64)
    p.add_argument("--max-wait-ms", type=float, default=2.0)
    p.add_argument("--device", default="auto")
    p.add_argument("--assets-dir", default="assets")
    p.add_argument("--config-file", default=None, help="TOML Config overriding flags")
    args = p.parse_args()
    if args.config_file:
        from ..config import Config

        cfg = Config.from_toml(args.config_file)
        args.host = cfg.serve.host
        args.port = cfg.serve.port
        args.max_batch_size = cfg.serve.max_batch_size
        args.max_wait_ms = cfg.serve.max_wait_ms
        args.assets_dir = cfg.bench.assets_dir

    import uvicorn

    from ..utils import load_data, load_model
    from .app import BatchKernelShapModel, create_app

    data = load_data(args.assets_dir)
    model_path = os.path.join(args.assets_dir, "predictor.pkl")
    if not os.path.exists(model_path):
        import subprocess
        import sys

        subprocess.run(
            [sys.executable, "scripts/fit_model.py", "--assets-dir", args.assets_dir],
            check=True,
        )
    clf = load_model(model_path)
    model = BatchKernelShapModel(
        clf,
        data.background,
        {"link": "logit", "seed": 0, "device": args.device},
        {"groups": data.groups, "group_names": data.group_names},
    )
    app = create_app(model, max_batch_size=args.max_batch_size,
                     max_wait_ms=args.max_wait_ms)
    uvicorn.run(app, host=args.host, port=args.port)


if __name__ == "__main__":
    main()
