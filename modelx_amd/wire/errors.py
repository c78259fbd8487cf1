"""Error vocabulary — OCI-distribution-style codes, wire-compatible with the
reference (pkg/errors/errors.go:11-44).

The HTTP body is ``{"code": ..., "message": ..., "detail": ...}`` with the
status carried on the response, not in the body.
"""
from __future__ import annotations

import json
from typing import Any, Dict


class ErrCode:
    """reference: pkg/errors/errors.go:11-31"""

    BLOB_UNKNOWN = "BLOB_UNKNOWN"
    BLOB_UPLOAD_INVALID = "BLOB_UPLOAD_INVALID"
    BLOB_UPLOAD_UNKNOWN = "BLOB_UPLOAD_UNKNOWN"
    DIGEST_INVALID = "DIGEST_INVALID"
    MANIFEST_BLOB_UNKNOWN = "MANIFEST_BLOB_UNKNOWN"
    MANIFEST_INVALID = "MANIFEST_INVALID"
    MANIFEST_UNKNOWN = "MANIFEST_UNKNOWN"
    NAME_INVALID = "NAME_INVALID"
    NAME_UNKNOWN = "NAME_UNKNOWN"
    SIZE_INVALID = "SIZE_INVALID"
    UNAUTHORIZED = "UNAUTHORIZED"
    DENIED = "DENIED"
    UNSUPPORTED = "UNSUPPORTED"
    TOO_MANY_REQUESTS = "TOOMANYREQUESTS"
    CONFIG_INVALID = "CONFIG_INVALID"
    INVALID_PARAMETER = "INVALID_PARAMETER"
    INDEX_UNKNOWN = "INDEX_UNKNOWN"
    UNKNOWN = "UNKNOWN"
    INTERNAL = "INTERNAL"


class ModelxError(Exception):
    """Carries an ErrorInfo; mirrors pkg/errors/errors.go:35-55."""

    def __init__(self, code: str, message: str, detail: str = "", http_status: int = 400):
        super().__init__(f"{code}: {message}")
        self.code = code
        self.message = message
        self.detail = detail
        self.http_status = http_status

    def to_dict(self) -> Dict[str, Any]:
        return {"code": self.code, "message": self.message, "detail": self.detail}

    def to_json(self) -> str:
        return json.dumps(self.to_dict(), separators=(",", ":"))

    @classmethod
    def from_dict(cls, d: Dict[str, Any], http_status: int = 400) -> "ModelxError":
        return cls(
            code=d.get("code", ErrCode.UNKNOWN) or ErrCode.UNKNOWN,
            message=d.get("message", "") or "",
            detail=d.get("detail", "") or "",
            http_status=http_status,
        )


def is_err_code(err: Exception, code: str) -> bool:
    return isinstance(err, ModelxError) and err.code == code


# Constructors mirroring pkg/errors/errors.go:57-107
def unauthorized(msg: str) -> ModelxError:
    return ModelxError(ErrCode.UNAUTHORIZED, msg, http_status=401)


def unsupported(msg: str) -> ModelxError:
    return ModelxError(ErrCode.UNSUPPORTED, msg, http_status=501)


def internal(msg: str) -> ModelxError:
    return ModelxError(ErrCode.INTERNAL, msg, http_status=500)


def digest_invalid(got: str) -> ModelxError:
    return ModelxError(ErrCode.DIGEST_INVALID, f"digest invalid: {got}", http_status=400)


def index_unknown(repository: str) -> ModelxError:
    return ModelxError(ErrCode.INDEX_UNKNOWN, f"index: {repository} not found", http_status=404)


def blob_unknown(digest: str) -> ModelxError:
    return ModelxError(ErrCode.BLOB_UNKNOWN, f"blob: {digest} not found", http_status=404)


def manifest_unknown(reference: str) -> ModelxError:
    return ModelxError(ErrCode.MANIFEST_UNKNOWN, f"manifest: {reference} not found", http_status=404)


def manifest_invalid(msg: str) -> ModelxError:
    return ModelxError(ErrCode.MANIFEST_INVALID, msg, http_status=400)


def content_type_invalid(got: str) -> ModelxError:
    return ModelxError(ErrCode.INVALID_PARAMETER, f"content type invalid: {got}", http_status=400)


def config_invalid(msg: str) -> ModelxError:
    return ModelxError(ErrCode.CONFIG_INVALID, msg, http_status=400)


def parameter_invalid(msg: str) -> ModelxError:
    return ModelxError(ErrCode.INVALID_PARAMETER, msg, http_status=400)
