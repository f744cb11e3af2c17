#!/usr/bin/env python3
"""Pre-download model weights onto a PV (reference
scripts/huggingface_downloader.py counterpart): fetches a repo's
safetensors + tokenizer into the layout
production_stack_amd.engine.weights.load_safetensors reads, so engine
pods with modelSpec.pvcStorage start without network access."""

import argparse
import sys


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("model_id", help="e.g. meta-llama/Llama-3.1-8B")
    ap.add_argument("output_dir")
    ap.add_argument("--revision", default=None)
    ap.add_argument("--token", default=None,
                    help="HF token (or set HF_TOKEN)")
    args = ap.parse_args()
    try:
        from huggingface_hub import snapshot_download
    except ImportError:
        print("huggingface_hub is required", file=sys.stderr)
        return 1
    path = snapshot_download(
        args.model_id,
        revision=args.revision,
        token=args.token,
        local_dir=args.output_dir,
        allow_patterns=["*.safetensors", "*.json", "tokenizer*",
                        "*.model"],
    )
    print(f"downloaded to {path}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
