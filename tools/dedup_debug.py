"""Deterministic dedup-path debugger (GPU). Pushes two blobs with identical
page-tiled tails, pulls the first (priming the chunk index), pulls the
second with dedup, and reports exactly which chunks mismatch and whether
they were gathered D2D or range-fetched."""
import hashlib
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                                "tests"))

import torch

from modelx_amd.client.gpu import GpuClient
from util_servers import start_modelxd_s3, start_s3d

CS = 128 << 10


def main():
    import tempfile

    work = tempfile.mkdtemp(prefix="dedup-dbg-")
    s3d = start_s3d(os.path.join(work, "s3"))
    mdx = start_modelxd_s3(s3d.url, redirect=True)
    try:
        g = GpuClient(mdx.url, device=0, num_slots=8, slot_bytes=16 << 20, dedup=True)
        page = torch.randint(0, 256, (1 << 20,), dtype=torch.uint8, device="cuda:0")
        blobs = {}
        for i in range(2):
            t = page.repeat(48).contiguous()  # 48 MiB
            torch.manual_seed(i)
            t[: 2 << 20] = torch.randint(0, 256, (2 << 20,), dtype=torch.uint8,
                                         device="cuda:0")
            g.push_from_gpu("dbg/dd", f"d{i}", {"blob.bin": t})
            blobs[f"d{i}"] = t
        g.clear_chunk_index()

        out0 = g.pull_to_gpu("dbg/dd", "d0")
        assert torch.equal(out0["blob.bin"], blobs["d0"]), "d0 plain pull wrong"
        print("d0 ok; dedup tensors held:", len(g._dedup_tensors))

        # manual replay of the dedup path for d1 with full introspection
        man = g.remote.get_manifest("dbg/dd", "d1")
        (desc,) = [d for d in man.blobs if d.name == "blob.bin"]
        expect = g._expected_leaves("dbg/dd", desc)
        assert expect is not None, "no leaves sidecar"
        dst = torch.empty(desc.size, dtype=torch.uint8, device="cuda:0")
        missing, dedup_bytes = g.engine.dedup_pull(expect, dst.data_ptr(), CS, desc.size)
        print(f"plan: {dedup_bytes} deduped bytes, missing ranges {missing}")
        url, headers = g._download_url("dbg/dd", desc)
        fetched = g._fetch_ranges(url, headers, dst.data_ptr(), missing)
        print("fetched", fetched)
        got = g.engine.sha256_chunk_leaves(dst.data_ptr(), desc.size, CS)
        nchunks = len(expect) // 32
        fetched_offs = set()
        for off, ln in missing:
            for o in range(off, off + ln, CS):
                fetched_offs.add(o)
        bad = []
        for i in range(nchunks):
            if got[i * 32 : (i + 1) * 32] != expect[i * 32 : (i + 1) * 32]:
                kind = "FETCHED" if i * CS in fetched_offs else "GATHERED"
                bad.append((i, kind))
        print(f"bad chunks: {len(bad)} of {nchunks}")
        for i, kind in bad[:20]:
            # compare against the reference tensor to see what landed
            ref = blobs["d1"][i * CS : (i + 1) * CS]
            gotb = dst[i * CS : (i + 1) * CS]
            eq_ref = torch.equal(gotb, ref)
            h = hashlib.sha256(bytes(gotb.cpu().numpy())).hexdigest()[:12]
            print(f"  chunk {i} [{kind}] equal_to_ref={eq_ref} landed_sha={h}")
            if not eq_ref:
                # does the landed chunk equal some OTHER offset of the blob?
                for j in range(max(0, i - 3), min(nchunks, i + 4)):
                    if torch.equal(gotb, blobs["d1"][j * CS : (j + 1) * CS]):
                        print(f"    == ref chunk {j}")
                        break
        if not bad:
            print("DEDUP PATH OK")
    finally:
        mdx.stop()
        s3d.stop()


if __name__ == "__main__":
    main()
