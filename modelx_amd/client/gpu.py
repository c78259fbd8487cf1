"""GPU data plane: S3 → HBM pulls and HBM → S3 pushes on MI355X.

Orchestrates the native pinned-ring engine (modelx_amd._core, built from
core/src/engine.cpp + core/hip/sha256.hip):

- ``pull_to_gpu``: manifest → presigned URL per blob → parallel ranged GETs
  into pinned slots → hipMemcpyAsync onto the device buffer → CDNA4 SHA-256
  chunk kernel verifies the chunked digest. Returns ``dict[name,
  torch.Tensor]`` (uint8) resident in HBM.
- ``push_from_gpu``: chunk-digest the device buffers on GPU, presigned
  (multi)part upload streamed D2H through the pinned ring, manifest PUT last.
- multi-GPU fan-out: rank 0 pulls, RCCL broadcast over xGMI per pipeline
  chunk via torch.distributed — see ``fanout.py`` (fanout_pull_broadcast /
  fanout_pull_sharded).

The native extension is REQUIRED on a GPU box: if HIP devices are visible and
the extension is missing, this module raises instead of silently falling back
to a CPU path.
"""
from __future__ import annotations

from datetime import datetime, timezone
from typing import Dict, List, Optional, Tuple

from ..wire import digest as dg
from ..wire import errors as er
from ..wire import types
from .registry import RegistryClient

DEFAULT_SLOT_BYTES = 64 << 20
DEFAULT_NUM_SLOTS = 16
DEFAULT_NUM_CONNS = 16  # measured: 16 conns/16 slots -> 17.3 vs 14.3 GiB/s at 8/10 (bench sweep)
DEFAULT_PART_BYTES = 256 << 20  # push part granularity (multipart)
DEFAULT_PUSH_PARALLEL = 6
# 128 KiB chunks put 65k+ SHA-256 chains in flight for multi-GiB blobs — the
# measured sweet spot on MI355X (1017 GiB/s vs 147 GiB/s at 1 MiB chunks,
# profiles/sha256_kernel.md)
DEFAULT_GPU_CHUNK = 128 << 10


def _core():
    try:
        from modelx_amd import _core as core
    except ImportError as e:  # pragma: no cover
        raise er.ModelxError(
            er.ErrCode.INTERNAL,
            "modelx_amd._core native extension not built "
            "(python setup_ext.py build_ext --inplace)",
        ) from e
    return core


def _tls_verify() -> bool:
    """requests verify flag for presigned-URL calls; MODELX_TLS_INSECURE=1
    (self-signed stores) matches the native engine's switch."""
    import os

    if os.environ.get("MODELX_TLS_INSECURE") == "1":
        import urllib3

        urllib3.disable_warnings()
        return False
    return True


def _signed_headers(part: dict) -> Dict[str, str]:
    out = {}
    for k, v in (part.get("signedHeader") or {}).items():
        out[k] = ",".join(v) if isinstance(v, list) else str(v)
    return out


class GpuClient:
    """Per-device transfer client. One GpuEngine (pinned ring + streams +
    hash stream) per instance."""

    def __init__(self, registry: str, authorization: str = "", device: int = 0,
                 num_slots: int = DEFAULT_NUM_SLOTS, slot_bytes: int = DEFAULT_SLOT_BYTES,
                 num_conns: int = DEFAULT_NUM_CONNS, dedup: bool = False):
        core = _core()
        if not core.hip_available():
            raise er.ModelxError(er.ErrCode.INTERNAL, "no HIP device visible")
        self.remote = RegistryClient(registry, authorization)
        self.device = device
        self.num_conns = num_conns
        self.slot_bytes = slot_bytes
        self.engine = core.GpuEngine(device=device, num_slots=num_slots,
                                     slot_bytes=slot_bytes, num_streams=4)
        self.last_stats: List[dict] = []
        # cross-blob chunk dedup (SURVEY.md §2.2 chunk_verify_dedup): an
        # HBM-resident hash table (core/hip/dedup.hip) maps leaf digests to
        # resident chunk addresses; a pull probes it on-device and gathers
        # hits D2D at HBM bandwidth instead of re-fetching from S3. Opt-in:
        # registered tensors are kept alive in _dedup_tensors.
        self.dedup = dedup
        self._dedup_tensors: List[object] = []
        self._dedup_registered = False

    # ----------------------------------------------------------- helpers --

    def _sync_producers(self) -> None:
        """Synchronize torch's streams before the engine READS a
        user-provided tensor. The engine runs on its own non-blocking HIP
        streams, which never implicitly order against torch's default
        stream — without this, digesting/pushing a tensor whose producing
        kernel (randint/repeat/copy) is still in flight reads torn bytes
        and permanently stores content that matches no digest. (Observed
        on hardware with 2.4 GiB generated tensors; small tensors usually
        win the race, which is what made it look intermittent.)"""
        import torch

        torch.cuda.synchronize(self.device)

    def _verify_device_digest(self, ptr: int, size: int, desc: types.Descriptor) -> None:
        """GPU chunk-digest the landed buffer and compare against the
        descriptor's chunked digest (or its chunk annotation)."""
        expect = None
        algo_cs = dg.algo_chunk_size(desc.digest.split(":", 1)[0]) if desc.digest else None
        if algo_cs:
            expect, cs = desc.digest, algo_cs
        else:
            note = desc.annotations.get(types.ANNOTATION_CHUNK_DIGEST, "")
            if note:
                cs = dg.algo_chunk_size(note.split(":", 1)[0]) or dg.DEFAULT_CHUNK_SIZE
                expect = note
        if expect is None:
            # canonical sha256 only (no chunk annotation): one sequential
            # chain — D2H through the pinned ring + SHA-NI on CPU, the fast
            # path for a single chain (a GPU lane is ~4x slower)
            digest = self.engine.sha256_canonical_device(ptr, size)
            got = "sha256:" + digest.hex()
            if got != desc.digest:
                raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                                     f"GPU digest mismatch for {desc.name}: {got}")
            return
        leaves = self.engine.sha256_chunk_leaves(ptr, size, cs)
        got = dg.root_from_leaf_bytes(leaves, cs, size)
        if got != expect:
            raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                                 f"GPU chunk digest mismatch for {desc.name}: {got} != {expect}")

    def _download_url(self, repository: str,
                      desc: types.Descriptor) -> Optional[Tuple[str, Dict[str, str]]]:
        """Presigned GET URL for a blob, or None when the registry runs
        without --enable-redirect (the caller degrades to streaming the
        bytes through the registry, pull.go:206-215 semantics)."""
        loc = self.remote.get_blob_location(repository, desc, "download")
        if loc is None:
            return None
        parts = loc.properties.get("parts") or []
        if not parts:
            raise er.ModelxError(er.ErrCode.UNKNOWN, "empty location parts")
        return parts[0]["url"], _signed_headers(parts[0])

    def _stream_via_registry(self, repository: str, desc: types.Descriptor,
                             tensor) -> None:
        """Fallback landing path for a redirect-less registry: stream the
        blob through the registry API into HBM in bounded pinned pieces.
        Slow (single stream, extra hop) but degrades instead of failing."""
        import time

        import torch

        t0 = time.monotonic()
        staging = torch.empty(self.slot_bytes, dtype=torch.uint8, pin_memory=True)
        fill = 0
        off = 0

        def flush():
            nonlocal fill, off
            if fill:
                tensor[off:off + fill].copy_(staging[:fill], non_blocking=False)
                off += fill
                fill = 0

        for chunk in self.remote.get_blob_content(repository, desc.digest):
            view = memoryview(chunk)
            while view:
                n = min(len(view), self.slot_bytes - fill)
                staging[fill:fill + n] = torch.frombuffer(bytearray(view[:n]),
                                                          dtype=torch.uint8)
                fill += n
                view = view[n:]
                if fill == self.slot_bytes:
                    flush()
        flush()
        if off != desc.size:
            raise er.ModelxError(er.ErrCode.UNKNOWN,
                                 f"registry stream truncated: {off} != {desc.size}")
        self.last_stats.append({"phase": "pull-registry-stream", "bytes": off,
                                "seconds": time.monotonic() - t0})

    # -------------------------------------------------------------- pull --

    @staticmethod
    def _plan_url(entry) -> Optional[Tuple[str, Dict[str, str]]]:
        """(url, headers) from a pull-plan blob entry, if it has one."""
        loc = (entry or {}).get("location") or {}
        parts = (loc.get("properties") or {}).get("parts") or []
        if not parts:
            return None
        return parts[0]["url"], _signed_headers(parts[0])

    @staticmethod
    def _plan_leaves(entry, desc: types.Descriptor) -> Optional[bytes]:
        """Inlined leaves from a pull-plan entry, VERIFIED against the
        descriptor's leaves-blob digest (the plan is untrusted input)."""
        b64 = (entry or {}).get("leaves64")
        if not b64:
            return None
        import base64

        try:
            data = base64.b64decode(b64)
        except Exception:
            return None
        ld = desc.annotations.get(types.ANNOTATION_LEAVES_BLOB, "")
        if not ld or dg.sha256_digest(data) != ld:
            return None
        return data

    def _expected_leaves(self, repository: str, desc: types.Descriptor) -> Optional[bytes]:
        """Fetch the leaves sidecar blob (per-chunk SHA-256 array) if the
        descriptor carries one."""
        ld = desc.annotations.get(types.ANNOTATION_LEAVES_BLOB, "")
        if not ld:
            return None
        try:
            data = b"".join(self.remote.get_blob_content(repository, ld))
        except er.ModelxError:
            return None
        if dg.sha256_digest(data) != ld:
            return None
        return data

    @staticmethod
    def _bad_chunk_ranges(got: bytes, expect: bytes, chunk_size: int,
                          size: int) -> List[Tuple[int, int]]:
        """Contiguous (offset, length) ranges of mismatching chunks."""
        n = len(expect) // 32
        bad = [i for i in range(n) if got[i * 32 : (i + 1) * 32] != expect[i * 32 : (i + 1) * 32]]
        ranges: List[Tuple[int, int]] = []
        for i in bad:
            off = i * chunk_size
            ln = min(chunk_size, size - off)
            if ranges and ranges[-1][0] + ranges[-1][1] == off:
                ranges[-1] = (ranges[-1][0], ranges[-1][1] + ln)
            else:
                ranges.append((off, ln))
        return ranges

    # ------------------------------------------------------ chunk dedup --

    def register_chunks(self, tensor, leaves: bytes, chunk_size: int) -> None:
        """Add a blob's chunks to the HBM dedup table (dedup.hip insert
        kernel); keeps the tensor alive while registered."""
        if not self.dedup:
            return
        size = tensor.numel() * tensor.element_size()
        self._dedup_tensors.append(tensor)
        self.engine.dedup_register(leaves, tensor.data_ptr(), chunk_size, size)
        self._dedup_registered = True

    def clear_chunk_index(self) -> None:
        if self._dedup_registered:
            self.engine.dedup_reset(0)
        self._dedup_tensors.clear()
        self._dedup_registered = False

    def _fetch_ranges(self, url: str, headers: Dict[str, str], ptr: int,
                      ranges: List[Tuple[int, int]]) -> int:
        if not ranges:
            return 0
        self.engine.pull_ranges_to_device(url, headers, ranges, ptr, self.num_conns)
        return sum(ln for _, ln in ranges)

    def pull_zstd_blob_to_device(self, repository: str, desc: types.Descriptor,
                                 verify: bool = True, plan_entry=None) -> "torch.Tensor":
        """Pull a +zstd blob: land the compressed stream in HBM (streaming
        SHA-256 verify over the stored bytes), decode every frame in its own
        workgroup (core/hip/zstd.hip), then GPU-verify the uncompressed
        chunked digest from the raw-digest annotation. The whole path —
        transfer, hash, inflate, re-hash — never leaves the GPU."""
        import time

        import torch

        comp = self.pull_blob_to_device(repository, desc, verify=verify,
                                        plan_entry=plan_entry)
        raw_size = int(desc.annotations.get(types.ANNOTATION_RAW_SIZE, 0) or 0)
        if not raw_size:
            raise er.ModelxError(er.ErrCode.UNSUPPORTED,
                                 f"+zstd blob {desc.name} lacks the raw-size annotation")
        out = torch.empty(max(raw_size, 1), dtype=torch.uint8, device=f"cuda:{self.device}")
        t0 = time.monotonic()
        try:
            got = self.engine.zstd_decompress_device(comp.data_ptr(), desc.size,
                                                     out.data_ptr(), out.numel())
        except RuntimeError as e:
            # diagnostic: persist the (digest-verified) compressed bytes so a
            # decode failure is analyzable offline; then retry once — a
            # second identical failure is deterministic (bad stored frame),
            # a pass means a device-side race to hunt. Opt-in via
            # MODELX_ZSTD_DUMP=<dir>: the blob can be gigabytes, so never
            # write it into an arbitrary cwd by default.
            import os as _os
            import sys

            dump_dir = _os.environ.get("MODELX_ZSTD_DUMP", "")
            if dump_dir:
                try:
                    _os.makedirs(dump_dir, exist_ok=True)
                    path = _os.path.join(
                        dump_dir,
                        f"zstd_decode_fail_{desc.digest.split(':')[-1][:12]}.zst")
                    with open(path, "wb") as f:
                        f.write(bytes(comp[: desc.size].cpu().numpy().tobytes()))
                    print(f"modelx: zstd decode failed ({e}); dumped {path}; retrying once",
                          file=sys.stderr)
                except Exception:
                    pass
            else:
                print(f"modelx: zstd decode failed ({e}); retrying once "
                      "(set MODELX_ZSTD_DUMP=<dir> to keep the blob)", file=sys.stderr)
            got = self.engine.zstd_decompress_device(comp.data_ptr(), desc.size,
                                                     out.data_ptr(), out.numel())
        self.last_stats.append({"phase": "pull-zstd-decompress", "bytes": got,
                                "seconds": time.monotonic() - t0})
        if raw_size and got != raw_size:
            raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                                 f"zstd size mismatch for {desc.name}: {got} != {raw_size}")
        out = out[:got]
        raw_digest = desc.annotations.get(types.ANNOTATION_RAW_DIGEST, "")
        if verify and raw_digest:
            cs = dg.algo_chunk_size(raw_digest.split(":", 1)[0]) or DEFAULT_GPU_CHUNK
            leaves = self.engine.sha256_chunk_leaves(out.data_ptr(), got, cs)
            root = dg.root_from_leaf_bytes(leaves, cs, got)
            if root != raw_digest:
                raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                                     f"uncompressed digest mismatch for {desc.name}")
        return out

    def pull_blob_to_device(self, repository: str, desc: types.Descriptor,
                            tensor=None, verify: bool = True,
                            resume: bool = False, plan_entry=None) -> "torch.Tensor":
        """Land a blob in HBM. With ``resume=True`` and an existing tensor,
        only chunks whose GPU hash mismatches the expected leaves are fetched
        (chunk-level resume/dedup — the reference resumes at whole-blob
        granularity only, pull.go:115-124). On digest mismatch after a full
        pull, only the BAD chunks are refetched (SURVEY.md §5 fault-injection
        requirement: a flipped byte must re-fetch the chunk, not the blob)."""
        import time

        import torch

        had_tensor = tensor is not None
        if tensor is None:
            tensor = torch.empty(desc.size, dtype=torch.uint8, device=f"cuda:{self.device}")
        assert tensor.numel() >= desc.size
        planned = self._plan_url(plan_entry)
        located = planned if planned else self._download_url(repository, desc)
        if located is None:
            # redirect-less registry: degrade to the streaming fallback
            for attempt in range(2):
                self._stream_via_registry(repository, desc, tensor)
                if not verify:
                    return tensor
                try:
                    self._verify_device_digest(tensor.data_ptr(), desc.size, desc)
                    return tensor
                except er.ModelxError:
                    if attempt:
                        raise
            return tensor
        url, headers = located

        def expected_leaves():
            return (self._plan_leaves(plan_entry, desc)
                    or self._expected_leaves(repository, desc))
        cs = int(desc.annotations.get(types.ANNOTATION_CHUNK_SIZE, 0) or 0) or (
            dg.algo_chunk_size(desc.digest.split(":", 1)[0]) or DEFAULT_GPU_CHUNK)

        if resume and had_tensor:
            expect = expected_leaves()
            if expect is not None:
                got = self.engine.sha256_chunk_leaves(tensor.data_ptr(), desc.size, cs)
                ranges = self._bad_chunk_ranges(got, expect, cs, desc.size)
                fetched = self._fetch_ranges(url, headers, tensor.data_ptr(), ranges)
                self.last_stats.append({"phase": "pull-resume", "bytes": fetched,
                                        "skipped": desc.size - fetched})
                if verify:
                    self._verify_device_digest(tensor.data_ptr(), desc.size, desc)
                return tensor
            # no leaves sidecar → fall through to a full pull

        if self.dedup and self._dedup_registered:
            # cross-blob dedup: probe the HBM table and gather resident
            # chunks D2D, fetch only the rest (SURVEY.md §2.2
            # chunk_verify_dedup; the reference dedups at whole-blob
            # granularity only, push.go:169-177)
            expect = expected_leaves()
            if expect is not None:
                t0 = time.monotonic()
                missing, dedup_bytes = self.engine.dedup_pull(
                    expect, tensor.data_ptr(), cs, desc.size)
                if dedup_bytes:
                    fetched = self._fetch_ranges(url, headers, tensor.data_ptr(), missing)
                    stats = {"phase": "pull-dedup", "bytes": fetched,
                             "dedup_bytes": dedup_bytes,
                             "seconds": time.monotonic() - t0}
                    self.last_stats.append(stats)
                    if verify:
                        # verify against the leaves; refetch any bad chunk
                        # (a stale/poisoned index entry must degrade to a
                        # fetch, never to a failure)
                        got = self.engine.sha256_chunk_leaves(tensor.data_ptr(),
                                                              desc.size, cs)
                        bad = self._bad_chunk_ranges(got, expect, cs, desc.size)
                        if bad:
                            import sys

                            fetched_offs = set()
                            for off, ln in missing:
                                for o in range(off, off + ln, cs):
                                    fetched_offs.add(o)
                            kinds = {"gathered": 0, "fetched": 0}
                            for off, ln in bad:
                                for o in range(off, off + ln, cs):
                                    kinds["fetched" if o in fetched_offs else "gathered"] += 1
                            print(f"modelx: dedup refetch {desc.name}: "
                                  f"{len(bad)} ranges, chunks by origin {kinds}",
                                  file=sys.stderr)
                            stats["refetched_bytes"] = self._fetch_ranges(
                                url, headers, tensor.data_ptr(), bad)
                            stats["refetched_ranges"] = len(bad)
                            got = self.engine.sha256_chunk_leaves(tensor.data_ptr(),
                                                                  desc.size, cs)
                        root = dg.root_from_leaf_bytes(got, cs, desc.size)
                        target = desc.annotations.get(types.ANNOTATION_CHUNK_DIGEST, "")
                        if dg.algo_chunk_size(desc.digest.split(":", 1)[0] if desc.digest
                                              else ""):
                            target = desc.digest
                        if target and root != target:
                            raise er.ModelxError(
                                er.ErrCode.DIGEST_INVALID,
                                f"GPU chunk digest mismatch for {desc.name}: {root}")
                    self.register_chunks(tensor, expect, cs)
                    return tensor

        # streaming-hash path: chunks are hashed on each slot's stream right
        # behind its H2D copy, so verification overlaps the transfer
        expect_digest = ""
        algo_cs = dg.algo_chunk_size(desc.digest.split(":", 1)[0]) if desc.digest else None
        if algo_cs:
            expect_digest = desc.digest
        elif desc.annotations.get(types.ANNOTATION_CHUNK_DIGEST, ""):
            expect_digest = desc.annotations[types.ANNOTATION_CHUNK_DIGEST]
        stream_hash = (verify and expect_digest
                       and self.slot_bytes % cs == 0 and desc.size > 0)
        t0 = time.monotonic()
        if stream_hash:
            stats, leaves = self.engine.pull_to_device_hashed(
                url, headers, desc.size, tensor.data_ptr(), self.num_conns, cs)
            stats["name"] = desc.name
            stats["phase"] = "pull-transfer-hashed"
            self.last_stats.append(stats)
            got = dg.root_from_leaf_bytes(leaves, cs, desc.size)
            if got == expect_digest:
                self.last_stats.append({"phase": "pull-verify", "bytes": desc.size,
                                        "seconds": time.monotonic() - t0 - stats["seconds"]})
                self.register_chunks(tensor, leaves, cs)
                return tensor
            # fall through to the refetch path below
        else:
            stats = self.engine.pull_to_device(url, headers, desc.size, tensor.data_ptr(),
                                               self.num_conns)
            stats["name"] = desc.name
            stats["phase"] = "pull-transfer"
            self.last_stats.append(stats)
        if not verify:
            return tensor
        t0 = time.monotonic()
        try:
            self._verify_device_digest(tensor.data_ptr(), desc.size, desc)
        except er.ModelxError as first_err:
            # chunk-level refetch before giving up
            expect = expected_leaves()
            if expect is None:
                raise er.ModelxError(
                    er.ErrCode.DIGEST_INVALID,
                    f"{first_err.message} [size={desc.size} media={desc.media_type} "
                    f"path=no-leaves-sidecar]") from first_err
            got = self.engine.sha256_chunk_leaves(tensor.data_ptr(), desc.size, cs)
            ranges = self._bad_chunk_ranges(got, expect, cs, desc.size)
            if not ranges:
                # landed leaves MATCH the sidecar yet the root digest
                # differs: the sidecar and the descriptor disagree — a
                # metadata-level inconsistency, not a transfer corruption
                raise er.ModelxError(
                    er.ErrCode.DIGEST_INVALID,
                    f"{first_err.message} [size={desc.size} media={desc.media_type} "
                    f"path=leaves-match-but-root-differs]") from first_err
            self._fetch_ranges(url, headers, tensor.data_ptr(), ranges)
            self.last_stats.append({"phase": "pull-chunk-refetch",
                                    "bytes": sum(r[1] for r in ranges)})
            try:
                self._verify_device_digest(tensor.data_ptr(), desc.size, desc)
            except er.ModelxError as second_err:
                raise er.ModelxError(
                    er.ErrCode.DIGEST_INVALID,
                    f"{second_err.message} [size={desc.size} media={desc.media_type} "
                    f"path=refetch-did-not-fix ranges={len(ranges)}]") from second_err
        self.last_stats.append({"phase": "pull-verify", "bytes": desc.size,
                                "seconds": time.monotonic() - t0})
        return tensor

    def _pull_zstd_batched(self, repository: str, jobs, verify: bool,
                           parallel: int) -> Dict[object, "torch.Tensor"]:
        """Many +zstd blobs in one orchestration pass: compressed bytes land
        (and stream-hash-verify) per blob, then ONE batched decode call and
        ONE batched leaf-digest call cover the whole set — the per-blob
        footer/table/sync round trips were the measured config-5 limiter
        (profiles/bench_shapes.md: 4.1 GiB/s effective vs 240+ GiB/s
        kernel). jobs = [(key, desc, plan_entry)]; returns {key: tensor}.

        Waves are capped at ~64 GiB of raw output so compressed+raw
        residency stays bounded on very large sets."""
        import time

        import torch

        wave_cap = self.ZSTD_WAVE_CAP
        if len(jobs) > 1:
            total_raw = sum(int(d.annotations.get(types.ANNOTATION_RAW_SIZE, 0) or 0)
                            for _, d, _ in jobs)
            if total_raw > wave_cap:
                out: Dict[object, "torch.Tensor"] = {}
                wave, acc = [], 0
                for job in jobs:
                    raw = int(job[1].annotations.get(types.ANNOTATION_RAW_SIZE, 0) or 0)
                    if wave and acc + raw > wave_cap:
                        out.update(self._pull_zstd_batched(repository, wave, verify,
                                                           parallel))
                        wave, acc = [], 0
                    wave.append(job)
                    acc += raw
                if wave:
                    out.update(self._pull_zstd_batched(repository, wave, verify,
                                                       parallel))
                return out

        def one(job):
            key, desc, entry = job
            return key, self.pull_blob_to_device(repository, desc, verify=verify,
                                                 plan_entry=entry)

        if parallel > 1 and len(jobs) > 1:
            from concurrent.futures import ThreadPoolExecutor

            with ThreadPoolExecutor(max_workers=parallel) as ex:
                comp = dict(ex.map(one, jobs))
        else:
            comp = dict(one(j) for j in jobs)
        outs: Dict[object, "torch.Tensor"] = {}
        items = []
        metas = []
        for key, desc, _ in jobs:
            raw_size = int(desc.annotations.get(types.ANNOTATION_RAW_SIZE, 0) or 0)
            if not raw_size:
                raise er.ModelxError(er.ErrCode.UNSUPPORTED,
                                     f"+zstd blob {desc.name} lacks the raw-size annotation")
            o = torch.empty(max(raw_size, 1), dtype=torch.uint8,
                            device=f"cuda:{self.device}")
            items.append((comp[key].data_ptr(), desc.size, o.data_ptr(), o.numel()))
            outs[key] = o
            metas.append((key, desc, raw_size))
        t0 = time.monotonic()
        sizes = self.engine.zstd_decompress_many(items)
        self.last_stats.append({"phase": "pull-zstd-decompress-batched",
                                "bytes": sum(sizes), "blobs": len(items),
                                "seconds": time.monotonic() - t0})
        for (key, desc, raw_size), got in zip(metas, sizes):
            if got != raw_size:
                raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                                     f"zstd size mismatch for {desc.name}: "
                                     f"{got} != {raw_size}")
            outs[key] = outs[key][:got]
        if verify:
            vitems, vmeta = [], []
            for key, desc, raw_size in metas:
                rd = desc.annotations.get(types.ANNOTATION_RAW_DIGEST, "")
                if not rd:
                    continue
                cs = dg.algo_chunk_size(rd.split(":", 1)[0]) or DEFAULT_GPU_CHUNK
                o = outs[key]
                vitems.append((o.data_ptr(), o.numel(), cs))
                vmeta.append((desc, cs, o.numel(), rd))
            if vitems:
                t0 = time.monotonic()
                leaves_list = self.engine.sha256_chunk_leaves_many(vitems)
                for (desc, cs, sz, rd), leaves in zip(vmeta, leaves_list):
                    if dg.root_from_leaf_bytes(leaves, cs, sz) != rd:
                        raise er.ModelxError(
                            er.ErrCode.DIGEST_INVALID,
                            f"uncompressed digest mismatch for {desc.name}")
                self.last_stats.append({"phase": "pull-zstd-raw-verify-batched",
                                        "bytes": sum(v[1] for v in vitems),
                                        "seconds": time.monotonic() - t0})
        return outs

    def pull_to_gpu(self, repository: str, version: str = "",
                    verify: bool = True, parallel: int = 1) -> Dict[str, "torch.Tensor"]:
        """Pull every file blob of a manifest into HBM. Directory (tar.gz)
        blobs are landed as raw archive bytes under their blob name.
        ``parallel`` > 1 pulls blobs concurrently (reentrant engine; right
        for many-shard manifests where per-blob latency would stack).
        Multiple +zstd blobs decode through the batched engine call."""
        # one round trip for manifest + presigns + leaves when the server
        # supports pull plans (measured: the per-blob control plane
        # dominated many-small-blob indexes — docs/roadmap.md)
        plan = self.remote.get_pull_plan(repository, version)
        if plan and plan.get("manifest"):
            manifest = types.Manifest.from_dict(plan["manifest"])
            plan_blobs = plan.get("blobs") or {}
        else:
            manifest = self.remote.get_manifest(repository, version)
            plan_blobs = {}
        descs = [d for d in manifest.blobs
                 if d.size and d.media_type != types.MEDIA_TYPE_MODEL_LEAVES]
        zstd_descs = [d for d in descs
                      if d.media_type == types.MEDIA_TYPE_MODEL_FILE_ZSTD]
        plain_descs = [d for d in descs
                       if d.media_type != types.MEDIA_TYPE_MODEL_FILE_ZSTD]

        out: Dict[str, "torch.Tensor"] = {}
        if len(zstd_descs) > 1:
            jobs = [(d.name, d, plan_blobs.get(d.digest)) for d in zstd_descs]
            out.update(self._pull_zstd_batched(repository, jobs, verify, parallel))
            zstd_descs = []

        def one(desc):
            entry = plan_blobs.get(desc.digest)
            if desc.media_type == types.MEDIA_TYPE_MODEL_FILE_ZSTD:
                return desc.name, self.pull_zstd_blob_to_device(repository, desc,
                                                                verify=verify,
                                                                plan_entry=entry)
            return desc.name, self.pull_blob_to_device(repository, desc, verify=verify,
                                                       plan_entry=entry)

        rest = plain_descs + zstd_descs
        if parallel > 1 and len(rest) > 1:
            from concurrent.futures import ThreadPoolExecutor

            with ThreadPoolExecutor(max_workers=parallel) as ex:
                out.update(dict(ex.map(one, rest)))
        else:
            out.update(dict(one(d) for d in rest))
        return out

    def pull_many(self, repository: str, versions, parallel: int = 6,
                  verify: bool = True) -> Dict[str, Dict[str, "torch.Tensor"]]:
        """Pull several versions concurrently. The native engine's pull path
        is reentrant (per-call range state, shared pinned-slot pool), so
        many small blobs overlap their HTTP round-trips — the limiter for
        mixed indexes (BASELINE config 5) is per-blob latency, not
        bandwidth. All +zstd blobs ACROSS the versions decode through one
        batched engine call (config 5 stores one small blob per version, so
        per-version batching alone would never engage)."""
        from concurrent.futures import ThreadPoolExecutor

        # one POST for all versions' plans when the server supports it
        try:
            batch = self.remote.get_pull_plans(repository, versions)
        except Exception:
            batch = None

        def info(v):
            plan = (batch or {}).get(v) or self.remote.get_pull_plan(repository, v)
            if plan and plan.get("manifest"):
                return v, types.Manifest.from_dict(plan["manifest"]), plan.get("blobs") or {}
            return v, self.remote.get_manifest(repository, v), {}

        with ThreadPoolExecutor(max_workers=parallel) as ex:
            infos = list(ex.map(info, versions))
        results: Dict[str, Dict[str, "torch.Tensor"]] = {v: {} for v in versions}
        zstd_jobs = []
        plain_jobs = []
        for v, manifest, plan_blobs in infos:
            for d in manifest.blobs:
                if not d.size or d.media_type == types.MEDIA_TYPE_MODEL_LEAVES:
                    continue
                entry = plan_blobs.get(d.digest)
                if d.media_type == types.MEDIA_TYPE_MODEL_FILE_ZSTD:
                    zstd_jobs.append(((v, d.name), d, entry))
                else:
                    plain_jobs.append((v, d, entry))
        if len(zstd_jobs) > 1:
            for (v, name), t in self._pull_zstd_batched(repository, zstd_jobs,
                                                        verify, parallel).items():
                results[v][name] = t
        elif zstd_jobs:
            (v, name), d, entry = zstd_jobs[0]
            results[v][name] = self.pull_zstd_blob_to_device(repository, d,
                                                             verify=verify,
                                                             plan_entry=entry)

        def one(job):
            v, d, entry = job
            return v, d.name, self.pull_blob_to_device(repository, d, verify=verify,
                                                       plan_entry=entry)

        if plain_jobs:
            with ThreadPoolExecutor(max_workers=parallel) as ex:
                for v, name, t in ex.map(one, plain_jobs):
                    results[v][name] = t
        return results

    # -------------------------------------------------- directory blobs --

    def _stream_targz_to_device(self, repository: str, desc: types.Descriptor):
        """Streamed landing of a tar.gz compat blob (reference
        pull.go:184-203 pipes download∥extract): presigned GET (or registry
        stream) → zlib inflate → bounded pinned staging → HBM, with the
        STORED bytes digest-verified on the fly — host memory stays
        O(staging), never O(blob). gzip is inherently sequential, so the
        inflate runs on CPU; the GPU-native directory format
        (MEDIA_TYPE_MODEL_DIRECTORY_TAR) skips this path entirely."""
        import time
        import zlib

        import torch

        t0 = time.monotonic()
        located = self._download_url(repository, desc)
        if located is not None:
            import requests

            url, headers = located
            resp = requests.get(url, headers=headers, stream=True,
                                verify=_tls_verify())
            resp.raise_for_status()
            source = resp.iter_content(chunk_size=1 << 20)
        else:
            source = self.remote.get_blob_content(repository, desc.digest)

        # verify the stored (compressed) bytes while streaming
        chunk_note = desc.annotations.get(types.ANNOTATION_CHUNK_DIGEST, "")
        cs = int(desc.annotations.get(types.ANNOTATION_CHUNK_SIZE, 0) or 0) or \
            dg.DEFAULT_CHUNK_SIZE
        hasher = dg.StreamingDigester(chunk_size=cs)

        inflater = zlib.decompressobj(16 + zlib.MAX_WBITS)  # gzip framing
        staging = torch.empty(self.slot_bytes, dtype=torch.uint8, pin_memory=True)
        fill = 0
        pieces: List["torch.Tensor"] = []

        def flush():
            nonlocal fill
            if fill:
                pieces.append(staging[:fill].to(f"cuda:{self.device}"))
                fill = 0

        def sink(data: bytes):
            nonlocal fill
            view = memoryview(data)
            while view:
                n = min(len(view), self.slot_bytes - fill)
                staging[fill:fill + n] = torch.frombuffer(bytearray(view[:n]),
                                                          dtype=torch.uint8)
                fill += n
                view = view[n:]
                if fill == self.slot_bytes:
                    flush()

        for piece in source:
            hasher.update(piece)
            sink(inflater.decompress(piece))
        sink(inflater.flush())
        flush()
        if hasher.total != desc.size:
            raise er.ModelxError(er.ErrCode.UNKNOWN,
                                 f"tar.gz stream truncated: {hasher.total} != {desc.size}")
        if desc.digest and desc.digest not in (hasher.canonical_digest(),
                                               hasher.chunk_digest()) and \
                (not chunk_note or chunk_note != hasher.chunk_digest()):
            raise er.ModelxError(er.ErrCode.DIGEST_INVALID,
                                 f"stored digest mismatch for {desc.name}")
        if not pieces:
            pieces = [torch.empty(0, dtype=torch.uint8, device=f"cuda:{self.device}")]
        archive = pieces[0] if len(pieces) == 1 else torch.cat(pieces)
        tar_len = archive.numel()
        self.last_stats.append({"phase": "pull-targz-stream", "bytes": tar_len,
                                "stored_bytes": desc.size,
                                "seconds": time.monotonic() - t0})
        return archive, tar_len

    def pull_dir_to_gpu(self, repository: str, desc: types.Descriptor
                        ) -> Dict[str, "torch.Tensor"]:
        """Land a directory blob and scatter its files into per-file HBM
        tensors with the CDNA4 tar kernels (core/hip/tar.hip). Plain-tar
        blobs (MEDIA_TYPE_MODEL_DIRECTORY_TAR) stay on-GPU end to end;
        tar+gz compat blobs are inflated on CPU first (gzip is sequential —
        the GPU-native path is the plain-tar format)."""
        import torch

        if desc.media_type == types.MEDIA_TYPE_MODEL_DIRECTORY_TAR:
            archive = self.pull_blob_to_device(repository, desc)
            tar_len = desc.size
        elif desc.media_type == types.MEDIA_TYPE_MODEL_DIRECTORY_TARGZ:
            archive, tar_len = self._stream_targz_to_device(repository, desc)
        else:
            raise er.ModelxError(er.ErrCode.UNSUPPORTED,
                                 f"not a directory blob: {desc.media_type}")
        entries = self.engine.tar_index(archive.data_ptr(), tar_len)
        out: Dict[str, "torch.Tensor"] = {}
        segs = []
        for e in entries:
            t = torch.empty(max(int(e["size"]), 1), dtype=torch.uint8,
                            device=f"cuda:{self.device}")
            if e["size"]:
                segs.append((int(e["offset"]), t.data_ptr(), int(e["size"])))
            out[e["name"]] = t[: int(e["size"])]
        if segs:
            self.engine.tar_scatter(archive.data_ptr(), segs)
        return out

    # -------------------------------------------------------------- push --

    def digest_device_blob(self, ptr: int, size: int,
                           chunk_size: int = DEFAULT_GPU_CHUNK) -> Tuple[str, str]:
        """(chunked_digest, chunk_digest_annotation) of device memory."""
        root, _ = self.digest_device_blob_with_leaves(ptr, size, chunk_size)
        return root, root

    def digest_device_blob_with_leaves(self, ptr: int, size: int,
                                       chunk_size: int = DEFAULT_GPU_CHUNK
                                       ) -> Tuple[str, bytes]:
        import time

        self._sync_producers()
        t0 = time.monotonic()
        leaves = self.engine.sha256_chunk_leaves(ptr, size, chunk_size)
        root = dg.root_from_leaf_bytes(leaves, chunk_size, size)
        self.last_stats.append({"phase": "gpu-digest", "bytes": size,
                                "seconds": time.monotonic() - t0})
        return root, leaves

    PART_RETRIES = 3  # reference: pkg/client/extension_s3.go:133-148
    ZSTD_WAVE_CAP = 64 << 30  # raw bytes per batched-decode wave

    def _upload_small(self, repository: str, desc: types.Descriptor, data: bytes) -> None:
        loc = self.remote.get_blob_location(repository, desc, "upload")
        if loc is not None:
            import requests

            p = (loc.properties.get("parts") or [{}])[0]
            for attempt in range(self.PART_RETRIES):
                try:
                    requests.request(p.get("method") or "PUT", p["url"],
                                     headers=_signed_headers(p), data=data,
                                     verify=_tls_verify()).raise_for_status()
                    return
                except Exception:
                    if attempt == self.PART_RETRIES - 1:
                        raise
        else:
            self.remote.upload_blob_content(repository, desc, data)

    def _push_part_retrying(self, part: dict, ptr: int, length: int) -> None:
        """One presigned part from device memory, retried on transport
        failure with a fresh connection (the engine drops broken sockets
        from its pool on error)."""
        for attempt in range(self.PART_RETRIES):
            try:
                self.engine.push_part_from_device(part["url"], part.get("method") or "PUT",
                                                  _signed_headers(part), ptr, length)
                return
            except RuntimeError:
                if attempt == self.PART_RETRIES - 1:
                    raise

    def push_blob_from_device(self, repository: str, desc: types.Descriptor, ptr: int,
                              part_bytes: int = DEFAULT_PART_BYTES,
                              parallel: int = DEFAULT_PUSH_PARALLEL) -> None:
        """Presigned (multi)part upload of device memory. HEAD-dedup first
        (push.go:169-177 semantics)."""
        import time

        self._sync_producers()
        t0 = time.monotonic()
        if self.remote.head_blob(repository, desc.digest):
            return
        size = desc.size
        nparts = max(1, (size + part_bytes - 1) // part_bytes)
        extra = {"part-count": str(nparts), "multipart": "true"} if nparts > 1 else None
        loc = self.remote.get_blob_location(repository, desc, "upload", extra=extra)
        if loc is None:
            # redirect-less registry: direct PUT through the registry
            # (reference pushBlob fallback, push.go:196-207), streamed D2H
            # in bounded slot-sized pieces. The reader carries a length so
            # requests sends Content-Length (the registry, like S3, takes
            # identity bodies, not chunked transfer encoding).
            engine, slot = self.engine, self.slot_bytes

            class _DeviceReader:
                def __init__(self):
                    self.len = size
                    self._off = 0
                    self._buf = b""

                def read(self, n=-1):
                    if n is None or n < 0:
                        n = size - self._off + len(self._buf)
                    while len(self._buf) < n and self._off < size:
                        take = min(slot, size - self._off)
                        self._buf += engine.read_device(ptr + self._off, take)
                        self._off += take
                    out, self._buf = self._buf[:n], self._buf[n:]
                    return out

            self.remote.upload_blob_content(repository, desc,
                                            _DeviceReader() if size else b"")
            self.last_stats.append({"phase": "push-registry-stream", "bytes": size,
                                    "seconds": time.monotonic() - t0})
            return
        parts = loc.properties.get("parts") or []
        ranges = []
        base = size // len(parts)
        off = 0
        for i in range(len(parts)):
            ln = base if i < len(parts) - 1 else size - off
            ranges.append((off, ln))
            off += ln
        if len(parts) == 1:
            self._push_part_retrying(parts[0], ptr, size)
        else:
            from concurrent.futures import ThreadPoolExecutor

            def send(i):
                o, ln = ranges[i]
                self._push_part_retrying(parts[i], ptr + o, ln)

            with ThreadPoolExecutor(max_workers=parallel) as pool:
                list(pool.map(send, range(len(parts))))
        self.last_stats.append({"phase": "push-upload", "bytes": size,
                                "seconds": time.monotonic() - t0, "parts": len(parts)})

    def push_from_gpu(self, repository: str, version: str,
                      tensors: Dict[str, "torch.Tensor"], config_yaml: str = "",
                      chunk_size: int = DEFAULT_GPU_CHUNK,
                      part_bytes: int = DEFAULT_PART_BYTES,
                      compress: str = "", digest_mode: str = "chunked") -> types.Manifest:
        """Digest on GPU → presigned multipart upload → manifest PUT last.
        With ``compress="zstd"`` each tensor is compressed on-GPU into a
        seekable multi-frame zstd blob (core/hip/zstd.hip) first; the
        descriptor digest covers the stored (compressed) bytes and the
        raw-digest/raw-size annotations carry the uncompressed identity.

        ``digest_mode="sha256"`` makes the main descriptor digest the
        wire-canonical plain sha256 (reference push.go:149-161 semantics —
        a stock Go modelx client digest-verifies the pulled blob with zero
        annotations consumed); the chunked digest then travels in the
        chunk-digest annotation, exactly like the CPU push path. The
        canonical chain streams D2H through the pinned ring onto the CPU's
        SHA-NI units. Default ``"chunked"`` keeps the all-GPU digest as the
        main digest (fastest; reference clients lose only verification)."""
        import torch

        if compress not in ("", "zstd"):
            raise er.ModelxError(er.ErrCode.UNSUPPORTED, f"unknown compression {compress!r}")
        if digest_mode not in ("chunked", "sha256"):
            raise er.ModelxError(er.ErrCode.UNSUPPORTED,
                                 f"unknown digest_mode {digest_mode!r}")
        manifest = types.Manifest(media_type=types.MEDIA_TYPE_MODEL_MANIFEST_JSON)
        now = datetime.now(timezone.utc)
        # config blob (small, CPU)
        cfg = config_yaml.encode() or b"description: pushed from GPU\n"
        cfg_digest = dg.sha256_digest(cfg)
        manifest.config = types.Descriptor(
            name="modelx.yaml", media_type=types.MEDIA_TYPE_MODEL_CONFIG_YAML,
            digest=cfg_digest, size=len(cfg), modified=now)
        if not self.remote.head_blob(repository, cfg_digest):
            self._upload_small(repository, manifest.config, cfg)
        if compress == "" and len(tensors) > 1:
            # many-blob fast path (BASELINE config 3 shape): ONE batched
            # leaf-digest call for all tensors, then blob uploads in
            # parallel threads; canonical mode digests via parallel CPU
            # SHA-NI chains (see below)
            import time
            from concurrent.futures import ThreadPoolExecutor

            self._sync_producers()
            items = list(tensors.items())
            ptrs = [t.data_ptr() for _, t in items]
            sizes = [t.numel() * t.element_size() for _, t in items]
            t0 = time.monotonic()
            leaves_list = self.engine.sha256_chunk_leaves_many(
                [(p, s, chunk_size) for p, s in zip(ptrs, sizes)])
            roots = [dg.root_from_leaf_bytes(lv, chunk_size, s)
                     for lv, s in zip(leaves_list, sizes)]
            self.last_stats.append({"phase": "gpu-digest-batched",
                                    "bytes": sum(sizes), "blobs": len(items),
                                    "seconds": time.monotonic() - t0})
            mains = list(roots)
            if digest_mode == "sha256":
                # parallel CPU SHA-NI chains, one per blob, over D2H.
                # Measured (config3, 64x1 GiB): the GPU multibuf kernel runs
                # 64 chains in ONE wave at ~16 MiB/s per latency-bound lane
                # (1.0 GiB/s aggregate) — a sequential chain has no
                # parallelism for the GPU to use, so the SHA-NI units win
                # by an order of magnitude across a few threads.
                import os as _os

                t0 = time.monotonic()
                workers = min(len(items), _os.cpu_count() or 8)
                with ThreadPoolExecutor(max_workers=workers) as ex:
                    digs = list(ex.map(
                        lambda a: self.engine.sha256_canonical_device(*a),
                        zip(ptrs, sizes)))
                mains = ["sha256:" + d.hex() for d in digs]
                self.last_stats.append({"phase": "push-canonical-digest-batched",
                                        "bytes": sum(sizes),
                                        "seconds": time.monotonic() - t0})
            descs = []
            for (name, t), root, main, lv, s in zip(items, roots, mains,
                                                    leaves_list, sizes):
                leaves_digest = dg.sha256_digest(lv)
                descs.append(types.Descriptor(
                    name=name, media_type=types.MEDIA_TYPE_MODEL_FILE,
                    digest=main, size=s, modified=now,
                    annotations={types.ANNOTATION_CHUNK_DIGEST: root,
                                 types.ANNOTATION_CHUNK_SIZE: str(chunk_size),
                                 types.ANNOTATION_LEAVES_BLOB: leaves_digest}))

            def upload(i):
                self.push_blob_from_device(repository, descs[i], ptrs[i],
                                           part_bytes=part_bytes)
                lv = leaves_list[i]
                ldesc = types.Descriptor(
                    name=items[i][0] + ".leaves",
                    media_type=types.MEDIA_TYPE_MODEL_LEAVES,
                    digest=descs[i].annotations[types.ANNOTATION_LEAVES_BLOB],
                    size=len(lv), modified=now)
                if not self.remote.head_blob(repository, ldesc.digest):
                    self._upload_small(repository, ldesc, lv)
                return ldesc

            with ThreadPoolExecutor(max_workers=DEFAULT_PUSH_PARALLEL) as ex:
                ldescs = list(ex.map(upload, range(len(items))))
            for (name, t), desc, lv, ldesc in zip(items, descs, leaves_list, ldescs):
                self.register_chunks(t, lv, chunk_size)
                manifest.blobs.append(desc)
                manifest.blobs.append(ldesc)
            manifest.blobs = types.sort_descriptors_by_name(manifest.blobs)
            self.remote.put_manifest(repository, version or "latest", manifest)
            return manifest
        for name, t in tensors.items():
            import time

            size = t.numel() * t.element_size()
            media_type = types.MEDIA_TYPE_MODEL_FILE
            extra_notes: Dict[str, str] = {}
            push_ptr, push_size = t.data_ptr(), size
            comp_keep = None  # keep the compressed tensor alive until pushed
            if compress == "zstd" and size:
                core = _core()
                self._sync_producers()
                raw_root, _ = self.digest_device_blob_with_leaves(t.data_ptr(), size,
                                                                  chunk_size)
                bound = core.zstd_compress_bound(size)
                comp_keep = torch.empty(bound, dtype=torch.uint8,
                                        device=f"cuda:{self.device}")
                t0 = time.monotonic()
                comp_size = self.engine.zstd_compress_device(
                    t.data_ptr(), size, 128 << 10, comp_keep.data_ptr(), bound)
                self.last_stats.append({"phase": "push-zstd-compress", "bytes": size,
                                        "seconds": time.monotonic() - t0,
                                        "compressed": comp_size})
                media_type = types.MEDIA_TYPE_MODEL_FILE_ZSTD
                extra_notes = {types.ANNOTATION_RAW_DIGEST: raw_root,
                               types.ANNOTATION_RAW_SIZE: str(size)}
                push_ptr, push_size = comp_keep.data_ptr(), comp_size
            root, leaves = self.digest_device_blob_with_leaves(push_ptr, push_size,
                                                               chunk_size)
            main_digest = root
            if digest_mode == "sha256":
                import time as _time

                t0 = _time.monotonic()
                canon = self.engine.sha256_canonical_device(push_ptr, push_size)
                main_digest = "sha256:" + canon.hex()
                self.last_stats.append({"phase": "push-canonical-digest",
                                        "bytes": push_size,
                                        "seconds": _time.monotonic() - t0})
            leaves_digest = dg.sha256_digest(leaves)
            desc = types.Descriptor(
                name=name, media_type=media_type, digest=main_digest, size=push_size,
                modified=now,
                annotations={types.ANNOTATION_CHUNK_DIGEST: root,
                             types.ANNOTATION_CHUNK_SIZE: str(chunk_size),
                             types.ANNOTATION_LEAVES_BLOB: leaves_digest,
                             **extra_notes})
            self.push_blob_from_device(repository, desc, push_ptr, part_bytes=part_bytes)
            if comp_keep is None:
                self.register_chunks(t, leaves, chunk_size)
            del comp_keep
            manifest.blobs.append(desc)
            # leaves sidecar: 32 B per chunk, enables chunk-level
            # resume/refetch/dedup on pull; listed in the manifest so GC
            # keeps it, with its own media type so clients can skip it
            ldesc = types.Descriptor(
                name=name + ".leaves", media_type=types.MEDIA_TYPE_MODEL_LEAVES,
                digest=leaves_digest, size=len(leaves), modified=now)
            if not self.remote.head_blob(repository, leaves_digest):
                self._upload_small(repository, ldesc, leaves)
            manifest.blobs.append(ldesc)
        manifest.blobs = types.sort_descriptors_by_name(manifest.blobs)
        self.remote.put_manifest(repository, version or "latest", manifest)
        return manifest
