"""creditcore — an MI355X-native online-inference serving framework.

Re-implements the capabilities of the reference MLOps POC
(`nfmoore/databricks-kubernetes-mlops-poc`) as a GPU-first serving stack:

- the reference trains a sklearn RandomForest credit-default classifier with
  drift (TabularDrift) and outlier (IForest) detection and serves it from a
  FastAPI ``POST /predict`` endpoint (reference: ``app/main.py``,
  ``databricks/src/01-train-model.ipynb``, ``02-register-model.ipynb``);
- creditcore keeps that feature schema, response contract and MLflow-pyfunc
  checkpoint layout, and rebuilds the execution engine natively for AMD
  Instinct MI355X (gfx950): hand-written HIP kernels for the scoring hot path
  (categorical encode + impute + forest traversal + isolation-forest scoring +
  drift statistics), micro-batched GPU serving, and data-parallel replicas
  across the 8 GPUs of one node with RCCL over xGMI.

Layout:
    schema.py    — request/response models (reference app/model.py:8-71)
    data.py      — synthetic UCI-shaped data + vocabularies
    train.py     — local trainer (reference 01-train-model.ipynb)
    registry.py  — MLflow pyfunc artifact layout (reference 02-register cell-12)
    pack.py      — pickles -> flat GPU buffers (tree SoA, vocabs, ECDFs)
    models/      — CPU reference detectors (forest/GBT, iforest, drift, linear)
    ops/         — HIP kernel wrappers (csrc/ extension) + CPU golden reference
    engine.py    — GPU scoring engine over the C++ ScoreSession (hipGraphs,
                   pinned slots, native JSON in/out)
    batching.py  — micro-batch request aggregator
    parallel.py  — RCCL weight broadcast + node-global drift sync
    serve.py     — FastAPI app: /predict /score /healthz /metrics /drift
    rawserve.py  — raw asyncio HTTP frontend (SO_REUSEPORT workers)
    dense.py     — 10M x 1k wide-tabular family (HBM-resident drift refs)
    pipeline.py  — local CD pipeline (train -> staging -> smoke -> prod)
"""

__version__ = "0.1.0"

FEATURE_COLUMNS = None  # populated lazily via creditcore.schema
