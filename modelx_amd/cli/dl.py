"""modelxdl — deployment-time downloader (Seldon/kubegems storage
initializer). Reference: cmd/modelxdl/modelxdl.go:42-98.

    modelxdl <modelx-uri> <dest-dir>          # land on local disk
    modelxdl <modelx-uri> --gpus 0,1,...      # land straight into HBM

Filters the manifest's blobs by the config's ``modelFiles`` first path
element (modelxdl.go:74-90), then pulls the selection.
"""
from __future__ import annotations

import argparse
import sys

import yaml

from ..client.reference import parse_reference
from ..config import ModelConfig
from ..wire import types


def select_blobs(manifest: types.Manifest, cfg: ModelConfig):
    """Filter blobs by modelFiles' first path element (modelxdl.go:74-90);
    empty modelFiles selects everything."""
    if not cfg.model_files:
        return list(manifest.blobs)
    roots = {mf.split("/", 1)[0] for mf in cfg.model_files}
    return [b for b in manifest.blobs if b.name in roots]


def main(argv=None) -> int:
    p = argparse.ArgumentParser(prog="modelxdl")
    p.add_argument("uri")
    p.add_argument("dest", nargs="?")
    p.add_argument("--gpus", default="",
                   help="comma-separated device ids: land shards straight into HBM")
    p.add_argument("--insecure", action="store_true")
    args = p.parse_args(argv)

    ref = parse_reference(args.uri)
    from ..client import Client

    client = Client(ref.registry, ref.authorization, insecure=args.insecure)
    manifest = client.get_manifest(ref.repository, ref.version)
    cfg = ModelConfig.from_dict(
        yaml.safe_load(client.get_config_content(ref.repository, ref.version)) or {})
    selection = select_blobs(manifest, cfg)

    if args.gpus:
        devices = [int(x) for x in args.gpus.split(",") if x != ""]
        from ..client.fanout import fanout_pull_single_process
        from ..wire import types as wt

        # leaves sidecars are pull metadata, not model content — don't
        # land them as HBM tensors
        gpu_sel = [b for b in selection
                   if b.media_type != wt.MEDIA_TYPE_MODEL_LEAVES]
        tensors = fanout_pull_single_process(ref, manifest, gpu_sel, devices)
        for dev, named in tensors.items():
            for name, t in named.items():
                print(f"cuda:{dev} {name}: {t.numel()} bytes in HBM")
        return 0

    if not args.dest:
        print("dest directory required (or --gpus)", file=sys.stderr)
        return 1
    client.puller.pull_blobs(ref.repository, selection + [manifest.config], args.dest)
    print(f"downloaded {len(selection)} blobs -> {args.dest}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
