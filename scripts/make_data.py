"""Generate + cache the synthetic Adult-shaped benchmark dataset
(reference scripts/process_adult_data.py pipeline, network-free)."""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--assets-dir", default="assets")
    args = p.parse_args()
    from distributedkernelshap_amd.utils import load_data

    data = load_data(args.assets_dir)
    print(
        f"dataset: X_train {data.X_train.shape}, X_test {data.X_test.shape}, "
        f"background {data.background.shape}, {len(data.groups)} groups"
    )


if __name__ == "__main__":
    main()
