"""modelx CLI — same verb surface as the reference
(cmd/modelx: init/push/pull/list/info/login/repo/completion, modelx.go:16-38)
plus MI355X-native extras (pull-gpu, gc).
"""
from __future__ import annotations

import argparse
import json
import os
import sys
from datetime import datetime, timezone

import yaml

from .._version import get as get_version
from ..config import MODEL_CONFIG_FILENAME, README_FILENAME, ModelConfig
from ..client.reference import parse_reference
from ..client.repos import RepoDetails, RepoManager
from ..client.units import human_size
from ..wire import errors as er


def _client_for(raw_ref: str, insecure: bool):
    ref = parse_reference(raw_ref)
    from ..client import Client

    return ref, Client(ref.registry, ref.authorization, insecure=insecure)


def cmd_init(args) -> int:
    """Scaffold modelx.yaml + README.md (reference: cmd/modelx/model/init.go:39-104)."""
    d = args.dir
    os.makedirs(d, exist_ok=True)
    cfg_path = os.path.join(d, MODEL_CONFIG_FILENAME)
    if os.path.exists(cfg_path) and not args.force:
        print(f"{cfg_path} already exists (use --force)", file=sys.stderr)
        return 1
    cfg = ModelConfig(
        description="modelx model",
        framework="pytorch",
        task="",
        tags=[],
        maintainers=[os.environ.get("USER", "unknown")],
        model_files=[],
        config={},
    )
    with open(cfg_path, "w") as f:
        f.write(cfg.to_yaml())
    readme = os.path.join(d, README_FILENAME)
    if not os.path.exists(readme):
        with open(readme, "w") as f:
            f.write(f"# {os.path.basename(os.path.abspath(d))}\n\nCreated by modelx init.\n")
    print(f"initialized {d}")
    return 0


def cmd_push(args) -> int:
    ref, client = _client_for(args.ref, args.insecure)
    if not ref.version:
        ref.version = "latest"
    manifest = client.push(ref.repository, ref.version, args.dir, configfile=args.config,
                           digest_mode=args.digest_mode, compress=args.compress)
    total = manifest.config.size + sum(b.size for b in manifest.blobs)
    print(f"pushed {ref.repository}@{ref.version}: {len(manifest.blobs)} blobs, "
          f"{human_size(total)}")
    return 0


def cmd_pull(args) -> int:
    ref, client = _client_for(args.ref, args.insecure)
    dest = args.dir or ref.repository.split("/")[-1]
    if args.gpu is not None:
        from ..client.gpu import GpuClient

        g = GpuClient(ref.registry, ref.authorization, device=args.gpu)
        tensors = g.pull_to_gpu(ref.repository, ref.version)
        for name, t in tensors.items():
            print(f"{name}: {human_size(t.numel())} in HBM (cuda:{args.gpu})")
        return 0
    client.pull(ref.repository, ref.version, dest)
    print(f"pulled {ref.repository}@{ref.version or 'latest'} -> {dest}")
    return 0


def cmd_list(args) -> int:
    """3 modes (reference: cmd/modelx/model/list.go:78-163)."""
    ref, client = _client_for(args.ref, args.insecure) if args.ref else (None, None)
    if ref is None or not ref.repository:
        # global index: list repositories
        if ref is None:
            print("usage: modelx list <registry-or-ref>", file=sys.stderr)
            return 1
        idx = client.get_global_index(args.search or "")
        for m in idx.manifests:
            print(m.name)
        return 0
    if not ref.version:
        idx = client.get_index(ref.repository, args.search or "")
        print(f"{'VERSION':<16}{'SIZE':<12}MODIFIED")
        for m in idx.manifests:
            mod = m.modified.strftime("%Y-%m-%d %H:%M:%S") if m.modified else ""
            print(f"{m.name:<16}{human_size(m.size):<12}{mod}")
        return 0
    manifest = client.get_manifest(ref.repository, ref.version)
    print(f"{'FILE':<32}{'TYPE':<12}{'SIZE':<12}DIGEST")
    for b in [manifest.config] + list(manifest.blobs):
        kind = {"application/vnd.modelx.model.directory.v1.tar+gz": "directory",
                "application/vnd.modelx.model.config.v1.yaml": "config"}.get(b.media_type, "file")
        print(f"{b.name:<32}{kind:<12}{human_size(b.size):<12}{b.digest[:23]}")
    return 0


def cmd_info(args) -> int:
    """Fetch + print the config blob (reference: info.go:47-65)."""
    ref, client = _client_for(args.ref, args.insecure)
    content = client.get_config_content(ref.repository, ref.version)
    sys.stdout.write(content.decode())
    return 0


def cmd_login(args) -> int:
    """Verify access then store the token (reference: login.go:51-62)."""
    ref, _ = _client_for(args.ref, args.insecure)
    token = args.token
    if not token:
        if sys.stdin.isatty():
            import getpass

            token = getpass.getpass("token: ")
        else:  # piped: modelx login NAME <<< "$TOKEN"
            token = sys.stdin.readline().strip()
    from ..client import Client

    client = Client(ref.registry, f"Bearer {token}" if token else "", insecure=args.insecure)
    client.ping()
    mgr = RepoManager()
    default_name = args.ref if "://" not in args.ref else \
        ref.registry.split("//", 1)[-1].split(":")[0]
    name = args.name or default_name.split("/", 1)[0]
    mgr.set(RepoDetails(name=name, url=ref.registry, token=token or ""))
    print(f"login succeeded: {name} -> {ref.registry}")
    return 0


def cmd_gc(args) -> int:
    ref, client = _client_for(args.ref, args.insecure)
    result = client.remote.garbage_collect(ref.repository)
    print(json.dumps(result))
    return 0


def cmd_version(args) -> int:
    print(json.dumps(get_version().to_dict(), indent=2))
    return 0


def cmd_repo(args) -> int:
    mgr = RepoManager()
    if args.repo_cmd == "add":
        # a ?token= in the URL moves into the stored token field
        # (reference: ?token= URI auth, cmd/modelx/model/reference.go:61-63)
        from urllib.parse import parse_qs

        url, token = args.url, args.token
        base, _, query = url.partition("?")
        if query:
            qtok = (parse_qs(query).get("token") or [""])[0]
            if qtok and not token:
                token = qtok
                url = base
        mgr.set(RepoDetails(name=args.name, url=url, token=token or ""))
        print(f"added repo {args.name} -> {url}")
    elif args.repo_cmd == "remove":
        if not mgr.remove(args.name):
            print(f"repo {args.name} not found", file=sys.stderr)
            return 1
        print(f"removed repo {args.name}")
    else:  # list
        print(f"{'NAME':<16}{'URL':<40}TOKEN")
        for r in mgr.list():
            print(f"{r.name:<16}{r.url:<40}{'***' if r.token else ''}")
    return 0


_COMPLETION_BASH = """
_modelx_completions() {
  local cur prev verbs
  cur="${COMP_WORDS[COMP_CWORD]}"
  verbs="init push pull list info login repo gc version completion"
  if [ $COMP_CWORD -eq 1 ]; then
    COMPREPLY=( $(compgen -W "$verbs" -- "$cur") )
  elif [ "${COMP_WORDS[1]}" = repo ]; then
    COMPREPLY=( $(compgen -W "add list remove" -- "$cur") )
  elif [[ "$cur" == */* ]]; then
    # live completion against the repo alias's index (reference: repo/list.go:42-106)
    COMPREPLY=( $(modelx list "${cur%%/*}" 2>/dev/null | sed "s|^|${cur%%/*}/|" | grep "^$cur") )
  fi
}
complete -F _modelx_completions modelx
"""


def cmd_completion(args) -> int:
    if args.shell in ("bash", "zsh"):
        print(_COMPLETION_BASH)
        return 0
    print(f"unsupported shell: {args.shell}", file=sys.stderr)
    return 1


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog="modelx",
                                description="modelx — MI355X-native model registry client")
    p.add_argument("--insecure", action="store_true", help="skip TLS verification")
    sub = p.add_subparsers(dest="cmd")

    sp = sub.add_parser("init", help="initialize a model directory")
    sp.add_argument("dir")
    sp.add_argument("--force", action="store_true")
    sp.set_defaults(fn=cmd_init)

    sp = sub.add_parser("push", help="push a model version")
    sp.add_argument("ref")
    sp.add_argument("dir", nargs="?", default=".")
    sp.add_argument("--config", default=MODEL_CONFIG_FILENAME)
    sp.add_argument("--digest-mode", choices=["sha256", "chunked"], default="sha256")
    sp.add_argument("--compress", choices=["", "zstd"], default="",
                    help="store file blobs zstd-compressed (seekable multi-frame, "
                         "GPU-parallel decode)")
    sp.set_defaults(fn=cmd_push)

    sp = sub.add_parser("pull", help="pull a model version")
    sp.add_argument("ref")
    sp.add_argument("dir", nargs="?")
    sp.add_argument("--gpu", type=int, default=None, metavar="DEV",
                    help="pull straight into HBM of GPU DEV")
    sp.set_defaults(fn=cmd_pull)

    sp = sub.add_parser("list", help="list repositories / versions / files")
    sp.add_argument("ref", nargs="?")
    sp.add_argument("--search", default="")
    sp.set_defaults(fn=cmd_list)

    sp = sub.add_parser("info", help="show a model's modelx.yaml")
    sp.add_argument("ref")
    sp.set_defaults(fn=cmd_info)

    sp = sub.add_parser("login", help="login to a registry")
    sp.add_argument("ref")
    sp.add_argument("--token", default="")
    sp.add_argument("--name", default="")
    sp.set_defaults(fn=cmd_login)

    sp = sub.add_parser("gc", help="garbage-collect unreferenced blobs of a repository")
    sp.add_argument("ref")
    sp.set_defaults(fn=cmd_gc)

    sp = sub.add_parser("version", help="print version info")
    sp.set_defaults(fn=cmd_version)

    sp = sub.add_parser("completion", help="shell completion script")
    sp.add_argument("shell", choices=["bash", "zsh", "fish", "powershell"])
    sp.set_defaults(fn=cmd_completion)

    sp = sub.add_parser("repo", help="repository alias management")
    rsub = sp.add_subparsers(dest="repo_cmd")
    ra = rsub.add_parser("add")
    ra.add_argument("name")
    ra.add_argument("url")
    ra.add_argument("--token", default="")
    rl = rsub.add_parser("list")
    rr = rsub.add_parser("remove")
    rr.add_argument("name")
    sp.set_defaults(fn=cmd_repo, repo_cmd="list")

    return p


def main(argv=None) -> int:
    parser = build_parser()
    args = parser.parse_args(argv)
    if not getattr(args, "cmd", None):
        parser.print_help()
        return 1
    try:
        return args.fn(args)
    except er.ModelxError as e:
        print(f"error: {e.code}: {e.message}", file=sys.stderr)
        return 1
    except (KeyError, ValueError, FileNotFoundError) as e:
        print(f"error: {e}", file=sys.stderr)
        return 1


if __name__ == "__main__":
    sys.exit(main())
