"""Wire-format tests: JSON schemas must match the reference Go structs
(pkg/types/types.go) byte-compatibly (field names; omitempty semantics)."""
import hashlib
import json
from datetime import datetime, timezone

import pytest

from modelx_amd.wire import digest as dg
from modelx_amd.wire import errors as er
from modelx_amd.wire import paths, types


class TestDescriptor:
    def test_minimal_roundtrip(self):
        d = types.Descriptor(name="weights.bin")
        j = d.to_dict()
        # name always present; modified always present (Go time.Time omitempty no-op)
        assert j["name"] == "weights.bin"
        assert j["modified"] == "0001-01-01T00:00:00Z"
        assert "mediaType" not in j and "digest" not in j and "size" not in j
        assert types.Descriptor.from_dict(j) == d

    def test_full_roundtrip(self):
        d = types.Descriptor(
            name="a/b.bin",
            media_type=types.MEDIA_TYPE_MODEL_FILE,
            digest="sha256:" + "ab" * 32,
            size=1234,
            mode=0o644,
            modified=datetime(2023, 5, 1, 12, 30, 15, 500000, tzinfo=timezone.utc),
            annotations={"filemode": "644"},
        )
        j = json.loads(types.dumps(d))
        assert j["mediaType"] == "application/vnd.modelx.model.file.v1"
        assert j["modified"] == "2023-05-01T12:30:15.5Z"
        d2 = types.Descriptor.from_dict(j)
        assert d2 == d

    def test_golden_reference_json_parses(self):
        # shaped like a real Go-marshalled descriptor
        golden = (
            '{"name":"model.safetensors","mediaType":"application/vnd.modelx.model.file.v1",'
            '"digest":"sha256:e3b0c44298fc1c149afbf4c8996fb92427ae41e4649b934ca495991b7852b855",'
            '"size":42,"mode":420,"modified":"2022-11-07T08:12:33.123456789Z"}'
        )
        d = types.Descriptor.from_dict(json.loads(golden))
        assert d.size == 42
        assert d.mode == 0o644
        assert d.modified.year == 2022
        assert d.modified.microsecond == 123456


class TestManifestIndex:
    def test_manifest_always_has_config_blobs_schema(self):
        m = types.Manifest()
        j = m.to_dict()
        assert j["schemaVersion"] == 1
        assert "config" in j and "blobs" in j
        assert types.Manifest.from_dict(j).schema_version == 1

    def test_index_roundtrip(self):
        idx = types.Index(
            media_type=types.MEDIA_TYPE_MODEL_INDEX_JSON,
            manifests=[types.Descriptor(name="v1"), types.Descriptor(name="v2")],
            annotations={"description": "demo"},
        )
        j = json.loads(types.dumps(idx))
        assert j["schemaVersion"] == 1
        assert [m["name"] for m in j["manifests"]] == ["v1", "v2"]
        assert types.Index.from_dict(j) == idx

    def test_sort_descriptors(self):
        descs = [types.Descriptor(name="b"), types.Descriptor(name="a")]
        assert [d.name for d in types.sort_descriptors_by_name(descs)] == ["a", "b"]

    def test_null_manifests_tolerated(self):
        # Go can emit "manifests":null for a nil slice
        idx = types.Index.from_dict(json.loads('{"schemaVersion":1,"manifests":null}'))
        assert idx.manifests == []


class TestBlobLocation:
    def test_presign_properties_schema(self):
        # Schema mirror of store_s3.go:228-233 / extension_s3.go:39-50
        loc = types.BlobLocation(
            provider="s3",
            purpose="upload",
            properties={
                "multipart": True,
                "uploadId": "xyz",
                "parts": [
                    {"url": "http://s3/x?sig=1", "method": "PUT", "partNumber": 1},
                    {"url": "http://s3/x?sig=2", "method": "PUT", "partNumber": 2},
                ],
            },
        )
        j = json.loads(types.dumps(loc))
        assert j["properties"]["parts"][1]["partNumber"] == 2
        assert types.BlobLocation.from_dict(j) == loc


class TestErrors:
    def test_error_body_schema(self):
        e = er.blob_unknown("sha256:" + "0" * 64)
        body = json.loads(e.to_json())
        assert set(body) == {"code", "message", "detail"}
        assert body["code"] == "BLOB_UNKNOWN"
        assert e.http_status == 404

    def test_codes_match_reference(self):
        # pkg/errors/errors.go:11-31
        assert er.ErrCode.TOO_MANY_REQUESTS == "TOOMANYREQUESTS"
        assert er.ErrCode.UNKNOWN == "UNKNOWN"
        assert er.unsupported("x").http_status == 501
        assert er.internal("x").http_status == 500
        assert er.unauthorized("x").http_status == 401

    def test_is_err_code(self):
        assert er.is_err_code(er.index_unknown("r"), er.ErrCode.INDEX_UNKNOWN)
        assert not er.is_err_code(ValueError("x"), er.ErrCode.INDEX_UNKNOWN)


class TestDigest:
    def test_sha256_matches_hashlib(self):
        data = b"hello modelx"
        assert dg.sha256_digest(data) == "sha256:" + hashlib.sha256(data).hexdigest()

    def test_empty_digest_constant(self):
        # push.go:25 EmptyFileDigiest
        assert dg.EMPTY_SHA256 == (
            "sha256:e3b0c44298fc1c149afbf4c8996fb92427ae41e4649b934ca495991b7852b855"
        )

    def test_parse_and_validate(self):
        algo, hexpart = dg.parse("sha256:" + "Ab" * 32)
        assert algo == "sha256" and hexpart == "ab" * 32
        assert dg.is_valid("sha256c1m:" + "0" * 64)
        assert not dg.is_valid("nohex")
        with pytest.raises(ValueError):
            dg.parse(":::")

    def test_chunked_algo_names(self):
        assert dg.chunked_algo_name(1 << 20) == "sha256c1m"
        assert dg.chunked_algo_name(64 << 10) == "sha256c64k"
        assert dg.algo_chunk_size("sha256c1m") == 1 << 20
        assert dg.algo_chunk_size("sha256c4k") == 4 << 10
        assert dg.algo_chunk_size("sha256") is None

    @pytest.mark.parametrize("n", [0, 1, 63, 64, 65, 1000, 4096, 10000])
    def test_chunked_digest_stream_equals_oneshot(self, n):
        data = bytes((i * 7 + 13) % 256 for i in range(n))
        cs = 1024
        one = dg.chunked_digest(data, cs)
        s = dg.StreamingDigester(chunk_size=cs)
        # odd-sized updates crossing chunk boundaries
        for off in range(0, n, 97):
            s.update(data[off : off + 97])
        assert s.chunk_digest() == one
        assert s.canonical_digest() == dg.sha256_digest(data)

    def test_chunked_digest_differs_on_chunk_swap(self):
        # swapping two equal-size chunks must change the root
        a, b = b"A" * 1024, b"B" * 1024
        assert dg.chunked_digest(a + b, 1024) != dg.chunked_digest(b + a, 1024)

    def test_chunked_digest_binds_length_and_chunksize(self):
        data = b"x" * 2048
        assert dg.chunked_digest(data, 1024) != dg.chunked_digest(data, 2048)


class TestPaths:
    def test_layout_matches_reference(self):
        # store.go:56-74
        d = "sha256:" + "ab" * 32
        assert paths.blob_digest_path("proj/name", d) == f"proj/name/blobs/sha256/{'ab' * 32}"
        assert paths.index_path("proj/name") == "proj/name/index.json"
        assert paths.manifest_path("proj/name", "v1") == "proj/name/manifests/v1"
        assert paths.S3_KEY_PREFIX == "registry"
