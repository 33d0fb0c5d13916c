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
    models/      — CPU reference detectors (forest, iforest, drift, linear)
    ops/         — HIP kernel wrappers (csrc/ extension)
    engine.py    — GPU scoring engine (HIP streams, pinned staging)
    batching.py  — micro-batch request aggregator
    parallel/    — RCCL replica group: weight broadcast + drift all-reduce
    serve.py     — FastAPI app: POST /predict (+ /score alias)
"""

__version__ = "0.1.0"

FEATURE_COLUMNS = None  # populated lazily via creditcore.schema
