from bodywork_mlops_demo_amd.store.base import ArtefactStore  # noqa: F401
from bodywork_mlops_demo_amd.store.local import LocalStore  # noqa: F401
from bodywork_mlops_demo_amd.store import contract  # noqa: F401


def open_store(uri: str | None = None) -> ArtefactStore:
    """Open an artefact store from a URI.

    ``s3://bucket`` opens the boto3-backed store (requires boto3 and AWS
    credentials, as the reference assumes — ``stage_1_train_model.py:61``);
    anything else is treated as a local directory path.  ``None`` uses the
    ``BODYWORK_AMD_STORE`` env var, falling back to ``./artefact-store``.
    """
    import os

    if uri is None:
        uri = os.environ.get("BODYWORK_AMD_STORE", "./artefact-store")
    if uri.startswith("s3://"):
        from bodywork_mlops_demo_amd.store.s3 import S3Store

        return S3Store(uri[len("s3://"):])
    return LocalStore(uri)
