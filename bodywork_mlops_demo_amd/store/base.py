"""Abstract artefact store — the framework's L1.

The reference talks to AWS S3 with a copy-pasted boto3 pattern in every
stage (list-objects by prefix → date-sort → get/put:
``stage_1_train_model.py:39-76``, ``stage_2:46-70``, ``stage_3:46-61``,
``stage_4:39-63``).  Here that pattern is one interface with local-POSIX
and S3 backends, so the whole DAG runs hermetically (tests, benchmarks)
or against real S3 (deployment parity).
"""
from __future__ import annotations

import io
from abc import ABC, abstractmethod
from datetime import date
from typing import Any

import numpy as np

from bodywork_mlops_demo_amd.store import contract


class ArtefactStore(ABC):
    """list/get/put over the 4-prefix, date-versioned key contract."""

    # -- primitive byte-level API ------------------------------------------
    @abstractmethod
    def list_keys(self, prefix: str) -> list[str]: ...

    @abstractmethod
    def get_bytes(self, key: str) -> bytes: ...

    @abstractmethod
    def put_bytes(self, key: str, data: bytes) -> None: ...

    @abstractmethod
    def exists(self, key: str) -> bool: ...

    def delete(self, key: str) -> None:  # optional
        raise NotImplementedError

    from contextlib import contextmanager as _cm

    @_cm
    def put_stream(self, key: str):
        """Streaming put: yields a writable binary file object.  Default
        buffers in memory; LocalStore overrides with a direct
        tmpfile+rename (no intermediate copy of large artefacts)."""
        bio = io.BytesIO()
        yield bio
        self.put_bytes(key, bio.getvalue())

    # -- contract-level helpers --------------------------------------------
    def latest(self, prefix: str) -> tuple[str, date]:
        """Key + date of the newest object under a prefix (stage_2:57-63)."""
        return contract.latest_key(self.list_keys(prefix))

    def all_by_date(self, prefix: str) -> list[tuple[str, date]]:
        """All objects under a prefix, time-ordered (stage_1:62-67)."""
        return contract.keys_by_date(self.list_keys(prefix))

    # -- dataset I/O --------------------------------------------------------
    # Canonical format: CSV with columns date,y,X (reference stage_3:42).
    # A binary .npz sidecar avoids CSV parse on the hot path for large N;
    # the CSV key remains the contract.

    def put_dataset(
        self,
        d: date,
        y: np.ndarray,
        X: np.ndarray,
        fmt: str = "csv",
    ) -> str:
        if fmt == "csv":
            key = contract.dataset_key(d, "csv")
            buf = io.StringIO()
            buf.write("date,y,X\n")
            ds = str(d)
            # vectorized CSV build — ~10x faster than pandas.to_csv for 1 col
            lines = np.char.add(
                np.char.add(f"{ds},", y.astype("U24")),
                np.char.add(",", X.astype("U24")),
            )
            buf.write("\n".join(lines.tolist()))
            buf.write("\n")
            self.put_bytes(key, buf.getvalue().encode())
        elif fmt in ("npy", "npz", "bin"):
            # two concatenated raw .npy records (y then X) streamed
            # straight to the destination: no zip/CRC pass and no
            # stack/BytesIO copies, so a 10M-row day persists at disk
            # bandwidth
            key = contract.dataset_key(d, "npy")
            with self.put_stream(key) as f:
                np.save(f, np.ascontiguousarray(y, dtype=np.float32))
                np.save(f, np.ascontiguousarray(X, dtype=np.float32))
        else:
            raise ValueError(f"unknown dataset format {fmt!r}")
        return key

    def get_dataset(self, key: str) -> tuple[np.ndarray, np.ndarray]:
        """Load (y, X) float32 arrays from a dataset artefact."""
        raw = self.get_bytes(key)
        if key.endswith(".npy"):
            bio = io.BytesIO(raw)
            first = np.load(bio)
            if first.ndim == 2:  # legacy single stacked [2, n] record
                return first[0], first[1]
            return first, np.load(bio)
        if key.endswith(".npz"):  # legacy binary artefacts
            z = np.load(io.BytesIO(raw))
            return z["y"], z["X"]
        # CSV: date,y,X header (stage_3:42); np.loadtxt is plenty here and
        # avoids a pandas dependency on the hot path
        arr = np.genfromtxt(
            io.BytesIO(raw), delimiter=",", skip_header=1, usecols=(1, 2),
            dtype=np.float64,
        )
        arr = np.atleast_2d(arr)
        return arr[:, 0].astype(np.float32), arr[:, 1].astype(np.float32)

    def get_all_datasets(self) -> tuple[np.ndarray, np.ndarray, date]:
        """Concatenate every dataset (time-ordered) — the reference's
        read-all-history training input (stage_1:59-71)."""
        pairs = self.all_by_date(contract.DATASETS_PREFIX)
        if not pairs:
            raise FileNotFoundError("no datasets in store")
        # one artefact per date: a day persisted in both formats (e.g. a
        # csv run resumed with persist_fmt='npy') must not be loaded twice
        # — prefer the binary record, which is the hot-path canonical form
        by_date: dict = {}
        for key, d in pairs:
            cur = by_date.get(d)
            if cur is None or (cur.endswith(".csv")
                               and not key.endswith(".csv")):
                by_date[d] = key
        pairs = sorted(((k, d) for d, k in by_date.items()),
                       key=lambda p: p[1])
        ys, Xs = [], []
        for key, _ in pairs:
            y, X = self.get_dataset(key)
            ys.append(y)
            Xs.append(X)
        return np.concatenate(ys), np.concatenate(Xs), pairs[-1][1]

    # -- model I/O (joblib-compatible: reference stage_1:111-125) -----------
    def put_model(self, model: Any, d: date) -> str:
        import joblib

        key = contract.model_key(d)
        bio = io.BytesIO()
        joblib.dump(model, bio)
        self.put_bytes(key, bio.getvalue())
        return key

    def get_latest_model(self) -> tuple[Any, date]:
        """Latest model by key-date (reference stage_2:46-70)."""
        import joblib

        key, d = self.latest(contract.MODELS_PREFIX)
        return joblib.load(io.BytesIO(self.get_bytes(key))), d

    # -- metric CSVs (schemas from stage_1:84-89 and stage_4:106-112) -------
    def put_metrics_csv(self, key: str, header: list[str], row: list) -> None:
        text = ",".join(header) + "\n" + ",".join(str(v) for v in row) + "\n"
        self.put_bytes(key, text.encode())

    def get_metrics_csv(self, key: str) -> dict:
        lines = self.get_bytes(key).decode().strip().splitlines()
        header = lines[0].split(",")
        vals = lines[1].split(",")
        return dict(zip(header, vals))
