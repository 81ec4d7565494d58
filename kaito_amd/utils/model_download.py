"""Model weight downloader — the payload of the ModelMirror download Job
(reference: pkg/modelmirror/download/job.go runs an HF snapshot download
with progress lines the controller samples)."""
from __future__ import annotations

import argparse
import os
import sys


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--model", required=True, help="HF repo id or preset name")
    p.add_argument("--dest", default="/weights")
    p.add_argument("--revision", default=None)
    p.add_argument("--token", default=os.environ.get("HF_TOKEN"))
    args = p.parse_args(argv)
    os.makedirs(args.dest, exist_ok=True)
    try:
        from huggingface_hub import snapshot_download
    except ImportError:
        print("huggingface_hub unavailable", file=sys.stderr)
        return 1
    # progress percent lines are sampled by the ModelMirror controller
    print("0% starting download", flush=True)
    path = snapshot_download(args.model, revision=args.revision,
                             token=args.token, local_dir=args.dest)
    print("100% download complete:", path, flush=True)
    with open(os.path.join(args.dest, ".download_complete"), "w") as f:
        f.write(args.model + "\n")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
