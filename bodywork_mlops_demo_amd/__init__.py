"""bodywork_mlops_demo_amd — an MI355X-native train-and-deploy MLOps framework.

A from-scratch re-design of the capabilities of AlexIoannides/bodywork-mlops-demo
(reference at /root/reference) for a single 8xMI355X node:

- the Kubernetes DAG that ``bodywork.yaml`` describes (reference
  ``bodywork.yaml:1-84``) becomes an in-process pipeline runner with HIP
  streams as DAG edges and one serving replica per GPU
  (:mod:`bodywork_mlops_demo_amd.pipeline`);
- the S3 artefact store and its 4-prefix date-versioned key contract
  (reference ``stage_1_train_model.py:39-76`` et al.) becomes
  :mod:`bodywork_mlops_demo_amd.store`;
- sklearn ``LinearRegression.fit`` / ``model.predict``
  (reference ``stage_1:105-106``, ``stage_2:78``) become hand-written
  CDNA4 HIP kernels in :mod:`bodywork_mlops_demo_amd.ops` (gfx950 MFMA /
  LDS-tiled, no CUDA shims, no Triton);
- the numpy synthetic-drift data generator (reference ``stage_3:28-43``)
  becomes an on-GPU philox kernel;
- multi-GPU training is data-parallel over RCCL/xGMI
  (:mod:`bodywork_mlops_demo_amd.parallel`).

Model artefacts stay joblib-compatible (reference ``stage_1:111-125``).
"""

__version__ = "0.1.0"

from bodywork_mlops_demo_amd.utils.logging import configure_logger  # noqa: F401
