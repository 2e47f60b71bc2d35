"""Copy app: fsspec-based cp (parity: torchx/apps/utils/copy_main.py)."""

import argparse

import fsspec


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--src", required=True)
    p.add_argument("--dst", required=True)
    args = p.parse_args()
    with fsspec.open(args.src, "rb") as r, fsspec.open(args.dst, "wb") as w:
        while True:
            chunk = r.read(1 << 20)
            if not chunk:
                break
            w.write(chunk)


if __name__ == "__main__":
    main()
