"""User configuration file (parity: python/scannerpy/config.py, which
reads ~/.scanner.toml with storage + network sections).

~/.scanner_amd.toml:

    [storage]
    type = "posix"            # only posix in this build (S3/GCS would
    db_path = "~/.scanner/db" # slot in behind StorageBackend)

    [network]
    master = "127.0.0.1"
    master_port = 5001

Environment override: SCANNER_AMD_CONFIG points at an alternate file.
"""
import os

from .common import ScannerException


def _load_toml(path):
    try:
        import tomli
    except ImportError:  # pragma: no cover
        import tomllib as tomli  # py311+
    with open(path, "rb") as f:
        return tomli.load(f)


class Config:
    def __init__(self, config_path=None, db_path=None, master=None):
        self.config_path = (config_path
                            or os.environ.get("SCANNER_AMD_CONFIG")
                            or os.path.expanduser("~/.scanner_amd.toml"))
        data = {}
        if os.path.isfile(self.config_path):
            try:
                data = _load_toml(self.config_path)
            except Exception as e:
                raise ScannerException(
                    f"failed to parse {self.config_path}: {e}")
        storage = data.get("storage", {})
        self.storage_type = storage.get("type", "posix")
        if self.storage_type not in ("posix", "s3", "gcs", "object"):
            raise ScannerException(
                f"unknown storage type '{self.storage_type}' (posix, or "
                "s3/gcs/object via the object-store backend, "
                "csrc/storage.h)")
        self.bucket = os.path.expanduser(storage.get("bucket", ""))
        if self.storage_type != "posix" and not self.bucket:
            raise ScannerException(
                f"storage type '{self.storage_type}' needs a bucket path")
        network = data.get("network", {})

        self.db_path = db_path or os.path.expanduser(
            storage.get("db_path", "~/.scanner_amd/db"))
        host = network.get("master", None)
        port = network.get("master_port", 5001)
        self.master_address = master or (
            f"{host}:{port}" if host is not None else None)
