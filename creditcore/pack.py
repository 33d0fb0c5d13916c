"""Pack trained models into flat GPU-uploadable buffers.

The reference's serving hot path is sklearn predict_proba + alibi-detect
drift/outlier scoring inside an MLflow pyfunc (reference 02-register cell-9).
creditcore converts those pickles ONCE at load time into flat SoA buffers laid
out for the MI355X HIP kernels, so serving never touches sklearn:

Node format (both forests), 16 B per node, loadable as one ``int4``:
    int32  feat    — virtual feature index; -1 => leaf
    f32    bits    — split threshold (internal) or leaf value (leaf)
    int32  left    — left child node index (x[feat] <= threshold)
    int32  right   — right child node index
Nodes of each tree are re-packed breadth-first so the hot top levels of every
tree are contiguous (cache-friendly traversal); all trees are concatenated
with a ``tree_offsets`` index.

Virtual feature space (classifier): the sklearn ColumnTransformer produces
[one-hot(9 categorical cols) | 14 numeric cols] (reference 01-train cell-6).
Rather than materialise the one-hot matrix, the kernel resolves a virtual
feature f through two small tables (kept in LDS):
    feat_col[f]  — source column (categorical col for one-hot, numeric col else)
    feat_code[f] — one-hot category code, or -1 for numeric
so  value(f) = (codes[row, col] == code) ? 1.0 : 0.0   for one-hot features
    value(f) = isnan(num) ? median[col] : num           for numeric features.

Isolation forest: leaf value = depth_edges + average_path_length(n_leaf) — the
per-tree path-length contribution sklearn computes at score time
(sklearn.ensemble._iforest); the anomaly score is then
2**(-sum/denom), denom = n_trees * average_path_length(max_samples_).

Drift reference state: per-numeric-feature sorted reference values (exact
two-sample K-S basis) and per-categorical-feature reference counts with one
extra "unseen" bin.
"""

from __future__ import annotations

from collections import deque
from dataclasses import dataclass, field

import numpy as np

from .schema import CATEGORICAL_FEATURES, MISSING_CATEGORY, NUMERIC_FEATURES

N_CAT = len(CATEGORICAL_FEATURES)
N_NUM = len(NUMERIC_FEATURES)

LEAF = -1


def _average_path_length(n: np.ndarray) -> np.ndarray:
    """sklearn.ensemble._iforest._average_path_length (re-derived: the
    expected path length of an unsuccessful BST search, Liu et al. 2008)."""
    n = np.asarray(n, dtype=np.float64)
    out = np.zeros_like(n)
    mask2 = n == 2
    maskg = n > 2
    out[mask2] = 1.0
    ng = n[maskg]
    out[maskg] = 2.0 * (np.log(ng - 1.0) + np.euler_gamma) - 2.0 * (ng - 1.0) / ng
    return out


def _pack_sklearn_tree(
    tree, leaf_values: np.ndarray, feature_remap: np.ndarray | None = None
) -> np.ndarray:
    """Repack one sklearn tree_ into BFS node-SoA [n,4] int32 (bits column
    stores f32 as raw bits)."""
    cl = tree.children_left
    cr = tree.children_right
    feat = tree.feature
    thr = tree.threshold
    n = len(cl)

    order = np.empty(n, dtype=np.int64)
    new_idx = np.full(n, -1, dtype=np.int64)
    q: deque[int] = deque([0])
    k = 0
    while q:
        i = q.popleft()
        order[k] = i
        new_idx[i] = k
        k += 1
        if cl[i] != -1:  # sklearn internal node (leaf children are -1)
            q.append(int(cl[i]))
            q.append(int(cr[i]))
    assert k == n, "disconnected sklearn tree"

    nodes = np.zeros((n, 4), dtype=np.int32)
    f32 = np.zeros(n, dtype=np.float32)
    for k in range(n):
        i = order[k]
        if cl[i] == -1:
            nodes[k, 0] = LEAF
            f32[k] = np.float32(leaf_values[i])
        else:
            f = int(feat[i])
            nodes[k, 0] = f if feature_remap is None else int(feature_remap[f])
            # sklearn predict casts X to float32 but keeps float64 thresholds;
            # ceil the threshold to the next float32 so that
            # {f32(x) <= thr32} == {f32(x) <= thr64} for every float32 x
            # (no float32 value lies in (thr64, ceil32(thr64))).
            t32 = np.float32(thr[i])
            if np.float64(t32) < thr[i]:
                t32 = np.nextafter(t32, np.float32(np.inf), dtype=np.float32)
            f32[k] = t32
            nodes[k, 2] = int(new_idx[cl[i]])
            nodes[k, 3] = int(new_idx[cr[i]])
    nodes[:, 1] = f32.view(np.int32)
    return nodes


def _concat_trees(tree_nodes: list[np.ndarray]) -> tuple[np.ndarray, np.ndarray]:
    offsets = np.zeros(len(tree_nodes) + 1, dtype=np.int32)
    for i, t in enumerate(tree_nodes):
        offsets[i + 1] = offsets[i] + len(t)
    return np.concatenate(tree_nodes, axis=0), offsets


@dataclass
class PackedModel:
    """Everything the GPU engine needs, as flat numpy arrays."""

    # feature encoding
    vocabs: list[list[str]]  # per categorical column, sklearn-OHE-sorted
    medians: np.ndarray  # f32[N_NUM]
    feat_col: np.ndarray  # i32[F_total]
    feat_code: np.ndarray  # i32[F_total]; -1 => numeric feature
    n_onehot: int

    # classifier forest
    cls_nodes: np.ndarray  # i32[n_nodes, 4]
    cls_tree_offsets: np.ndarray  # i32[T+1]

    # isolation forest (features index numeric columns directly)
    if_nodes: np.ndarray
    if_tree_offsets: np.ndarray
    if_denom: float  # n_trees * average_path_length(max_samples_)
    if_offset: float  # sklearn decision_function offset_ (-0.5)
    if_threshold: float  # alibi outlier threshold (0.95)

    # drift reference state
    n_ref: int
    drift_p_val: float
    ref_sorted: np.ndarray  # f32[sum n_ref per num feature] (equal n_ref each)
    ref_sorted_offsets: np.ndarray  # i32[N_NUM+1]
    ref_cat_counts: np.ndarray  # i32[sum (vocab+1)]
    ref_cat_offsets: np.ndarray  # i32[N_CAT+1]

    # classifier finalize: 0 = random forest (proba = leaf-fraction mean);
    # 1 = gradient-boosted (proba = sigmoid(sum of lr-scaled leaves + prior))
    cls_kind: int = 0
    cls_bias: float = 0.0

    # optional linear scorer
    lin_weight: np.ndarray | None = None  # f32[F_total]
    lin_bias: float = 0.0

    meta: dict = field(default_factory=dict)

    @property
    def n_features_total(self) -> int:
        return int(self.n_onehot + N_NUM)

    @property
    def cls_n_trees(self) -> int:
        return len(self.cls_tree_offsets) - 1

    @property
    def if_n_trees(self) -> int:
        return len(self.if_tree_offsets) - 1

    # -- persistence (fast serving start without sklearn unpickling) --------
    def save(self, path: str) -> None:
        np.savez_compressed(
            path,
            vocabs=np.asarray(
                ["\x00".join(v) for v in self.vocabs], dtype=object
            ),
            medians=self.medians,
            feat_col=self.feat_col,
            feat_code=self.feat_code,
            n_onehot=self.n_onehot,
            cls_nodes=self.cls_nodes,
            cls_tree_offsets=self.cls_tree_offsets,
            cls_kind=self.cls_kind,
            cls_bias=self.cls_bias,
            if_nodes=self.if_nodes,
            if_tree_offsets=self.if_tree_offsets,
            if_denom=self.if_denom,
            if_offset=self.if_offset,
            if_threshold=self.if_threshold,
            n_ref=self.n_ref,
            drift_p_val=self.drift_p_val,
            ref_sorted=self.ref_sorted,
            ref_sorted_offsets=self.ref_sorted_offsets,
            ref_cat_counts=self.ref_cat_counts,
            ref_cat_offsets=self.ref_cat_offsets,
            lin_weight=self.lin_weight if self.lin_weight is not None else np.zeros(0),
            lin_bias=self.lin_bias,
            allow_pickle=True,
        )

    @classmethod
    def load(cls, path: str) -> "PackedModel":
        z = np.load(path, allow_pickle=True)
        lw = z["lin_weight"]
        return cls(
            vocabs=[s.split("\x00") for s in z["vocabs"].tolist()],
            medians=z["medians"],
            feat_col=z["feat_col"],
            feat_code=z["feat_code"],
            n_onehot=int(z["n_onehot"]),
            cls_nodes=z["cls_nodes"],
            cls_tree_offsets=z["cls_tree_offsets"],
            cls_kind=int(z["cls_kind"]) if "cls_kind" in z else 0,
            cls_bias=float(z["cls_bias"]) if "cls_bias" in z else 0.0,
            if_nodes=z["if_nodes"],
            if_tree_offsets=z["if_tree_offsets"],
            if_denom=float(z["if_denom"]),
            if_offset=float(z["if_offset"]),
            if_threshold=float(z["if_threshold"]),
            n_ref=int(z["n_ref"]),
            drift_p_val=float(z["drift_p_val"]),
            ref_sorted=z["ref_sorted"],
            ref_sorted_offsets=z["ref_sorted_offsets"],
            ref_cat_counts=z["ref_cat_counts"],
            ref_cat_offsets=z["ref_cat_offsets"],
            lin_weight=lw if lw.size else None,
            lin_bias=float(z["lin_bias"]),
        )


def pack_classifier_pipeline(pipeline) -> dict:
    """Extract vocabularies, medians and the BFS node-SoA forest from the
    sklearn pipeline built by make_classifier_pipeline. Supports the
    reference's RandomForestClassifier (leaf-fraction mean) and
    GradientBoostingClassifier (lr-scaled regression leaves summed into a
    logit on top of the prior)."""
    pre = pipeline.named_steps["preprocessor"]
    clf = pipeline.named_steps["classifier"]

    cat_pipe = pre.named_transformers_["categorical"]
    num_pipe = pre.named_transformers_["numeric"]
    ohe = cat_pipe.named_steps["ohe"]
    vocabs = [list(map(str, c)) for c in ohe.categories_]
    medians = np.asarray(num_pipe.named_steps["imputer"].statistics_, dtype=np.float32)

    # virtual feature map: [all one-hot features | numeric features]
    feat_col, feat_code = [], []
    for col, vocab in enumerate(vocabs):
        for code in range(len(vocab)):
            feat_col.append(col)
            feat_code.append(code)
    n_onehot = len(feat_col)
    for col in range(N_NUM):
        feat_col.append(col)
        feat_code.append(-1)

    trees = []
    cls_kind, cls_bias = 0, 0.0
    if hasattr(clf, "loss_") or type(clf).__name__ == "GradientBoostingClassifier":
        cls_kind = 1
        lr = float(clf.learning_rate)
        # binary log-loss prior (raw score of the init estimator)
        import sklearn.dummy

        init = clf.init_
        if isinstance(init, str) and init == "zero":
            cls_bias = 0.0
        elif isinstance(init, sklearn.dummy.DummyClassifier):
            p1 = float(np.clip(init.class_prior_[1], 1e-12, 1 - 1e-12))
            cls_bias = float(np.log(p1 / (1.0 - p1)))
        else:  # pragma: no cover - custom init estimators
            raise ValueError("unsupported GBT init estimator")
        for est in clf.estimators_[:, 0]:
            t = est.tree_
            leaf_raw = np.asarray(t.value, dtype=np.float64)[:, 0, 0] * lr
            trees.append(_pack_sklearn_tree(t, leaf_raw))
    else:
        for est in clf.estimators_:
            t = est.tree_
            value = np.asarray(t.value, dtype=np.float64)  # (n_nodes, 1, 2)
            sums = value[:, 0, :].sum(axis=1)
            sums[sums == 0] = 1.0
            leaf_p1 = value[:, 0, 1] / sums  # fraction of class 1 (normalised
            # either way: sklearn >=1.4 already stores fractions)
            trees.append(_pack_sklearn_tree(t, leaf_p1))
    nodes, offsets = _concat_trees(trees)

    return {
        "vocabs": vocabs,
        "medians": medians,
        "feat_col": np.asarray(feat_col, dtype=np.int32),
        "feat_code": np.asarray(feat_code, dtype=np.int32),
        "n_onehot": n_onehot,
        "cls_nodes": nodes,
        "cls_tree_offsets": offsets,
        "cls_kind": cls_kind,
        "cls_bias": cls_bias,
    }


def pack_isolation_forest(detector) -> dict:
    """Pack the IForestDetector's sklearn IsolationForest into node-SoA with
    per-leaf path-length contributions (depth + average_path_length(n_leaf))."""
    iso = detector.isolationforest
    trees = []
    for est, feats in zip(iso.estimators_, iso.estimators_features_):
        t = est.tree_
        n = t.node_count
        # depth of each node in edges
        depth = np.zeros(n, dtype=np.int64)
        cl, cr = t.children_left, t.children_right
        stack = [(0, 0)]
        while stack:
            i, d = stack.pop()
            depth[i] = d
            if cl[i] != -1:
                stack.append((int(cl[i]), d + 1))
                stack.append((int(cr[i]), d + 1))
        leaf_val = depth + _average_path_length(t.n_node_samples)
        remap = np.asarray(feats, dtype=np.int64)  # subsampled feature ids
        trees.append(_pack_sklearn_tree(t, leaf_val, feature_remap=remap))
    nodes, offsets = _concat_trees(trees)
    denom = float(
        len(iso.estimators_) * _average_path_length(np.asarray([iso.max_samples_]))[0]
    )
    return {
        "if_nodes": nodes,
        "if_tree_offsets": offsets,
        "if_denom": denom,
        "if_offset": float(iso.offset_),
        "if_threshold": float(detector.threshold),
    }


def pack_drift(detector, vocabs: list[list[str]]) -> dict:
    """Pack the TabularDriftDetector reference state.

    Categorical reference counts are re-binned onto the classifier's OHE
    vocabulary (plus one trailing "unseen" bin) so the GPU encode produces one
    code per value usable by both the classifier and the drift histogram.
    """
    n_num = len(detector.numeric_idx)
    assert n_num == N_NUM
    ref_sorted = [
        np.asarray(detector.ref_sorted[i], dtype=np.float32) for i in detector.numeric_idx
    ]
    rs_off = np.zeros(N_NUM + 1, dtype=np.int32)
    for i, r in enumerate(ref_sorted):
        rs_off[i + 1] = rs_off[i] + len(r)

    cat_counts, cat_off = [], np.zeros(N_CAT + 1, dtype=np.int32)
    for j, i in enumerate(detector.categorical_idx):
        vocab = vocabs[j]
        idx = {c: k for k, c in enumerate(vocab)}
        counts = np.zeros(len(vocab) + 1, dtype=np.int32)  # +1 unseen bin
        for cat, cnt in zip(detector.categories[i], detector.ref_counts[i]):
            counts[idx.get(str(cat), len(vocab))] += int(cnt)
        cat_counts.append(counts)
        cat_off[j + 1] = cat_off[j] + len(counts)

    return {
        "n_ref": int(detector.n_ref),
        "drift_p_val": float(detector.p_val),
        "ref_sorted": np.concatenate(ref_sorted),
        "ref_sorted_offsets": rs_off,
        "ref_cat_counts": np.concatenate(cat_counts),
        "ref_cat_offsets": cat_off,
    }


def pack_pyfunc_dir(model_dir: str) -> PackedModel:
    """Load an MLflow pyfunc model dir (registry layout) and pack everything."""
    from . import registry

    loaded = registry.load_pyfunc_model(model_dir)
    m = loaded.python_model
    c = pack_classifier_pipeline(m.classifier)
    o = pack_isolation_forest(m.outliers)
    d = pack_drift(m.drift, c["vocabs"])
    return PackedModel(**c, **o, **d)


# ---------------------------------------------------------------------------
# Host-side request encoding (strings -> codes; floats pass through)
# ---------------------------------------------------------------------------


def encode_batch(records_or_df, vocabs: list[list[str]]) -> tuple[np.ndarray, np.ndarray]:
    """Encode a request batch into (codes int16 [B, 9], nums float32 [B, 14]).

    code = index into the column's OHE vocabulary; unknown category -> -1
    (one-hot all-zero, matching OneHotEncoder(handle_unknown="ignore")), and
    missing -> the MISSING_CATEGORY code if it is in-vocab, else -1.
    Numeric NaNs pass through (imputed to the median inside the scorer).

    The list-of-dicts path (the serving hot path: parsed request bodies)
    avoids pandas entirely — per-column C-level transposes + dict lookups.
    """
    if not isinstance(records_or_df, list):
        return _encode_df(records_or_df, vocabs)
    recs: list[dict] = records_or_df
    ext = _native_encoder()
    if ext is not None:
        codes, nums = ext.encode_records(
            recs, vocabs, CATEGORICAL_FEATURES, NUMERIC_FEATURES, MISSING_CATEGORY
        )
        return codes, nums
    return _encode_records_py(recs, vocabs)


def _native_encoder():
    from .ops import gpu

    return gpu._ext if gpu.available() else None


def _encode_records_py(recs: list, vocabs: list[list[str]]) -> tuple[np.ndarray, np.ndarray]:
    """Pure-Python fallback for the native encode_records (identical
    semantics; used only when the extension isn't built)."""
    from itertools import repeat
    from operator import itemgetter

    b = len(recs)
    codes = np.empty((b, N_CAT), dtype=np.int16)
    try:
        # serving fast path: complete records (pydantic fills defaults) —
        # C-level itemgetter transpose + map(dict.get, col, repeat(-1))
        cat_rows = list(map(itemgetter(*CATEGORICAL_FEATURES), recs))
        num_rows = list(map(itemgetter(*NUMERIC_FEATURES), recs))
    except KeyError:
        cat_rows = [tuple(r.get(c, MISSING_CATEGORY) for c in CATEGORICAL_FEATURES) for r in recs]
        num_rows = [tuple(r.get(c) for c in NUMERIC_FEATURES) for r in recs]
    for j, col in enumerate(zip(*cat_rows)):
        vmap = _vocab_map(tuple(vocabs[j]))
        codes[:, j] = np.fromiter(
            map(vmap.get, col, repeat(np.int16(-1))), dtype=np.int16, count=b
        )
    nums = np.array(num_rows, dtype=np.float32).reshape(b, N_NUM)
    return codes, nums


_VOCAB_MAP_CACHE: dict = {}


def _vocab_map(vocab: tuple) -> dict:
    """code map for one categorical column; None/NaN resolve like the
    MISSING_CATEGORY constant (SimpleImputer fill_value='missing')."""
    m = _VOCAB_MAP_CACHE.get(vocab)
    if m is None:
        m = {v: np.int16(k) for k, v in enumerate(vocab)}
        missing_code = m.get(MISSING_CATEGORY, np.int16(-1))
        m[None] = missing_code
        m[np.nan] = missing_code
        _VOCAB_MAP_CACHE[vocab] = m
    return m


def _encode_df(df, vocabs: list[list[str]]) -> tuple[np.ndarray, np.ndarray]:
    b = len(df)
    codes = np.full((b, N_CAT), -1, dtype=np.int16)
    for j, col in enumerate(CATEGORICAL_FEATURES):
        vocab = np.asarray(vocabs[j], dtype=object)
        vals = df[col].astype(object).fillna(MISSING_CATEGORY).astype(str).to_numpy()
        pos = np.searchsorted(vocab, vals)
        pos = np.clip(pos, 0, len(vocab) - 1)
        hit = vocab[pos] == vals
        codes[:, j] = np.where(hit, pos, -1).astype(np.int16)
    nums = df[NUMERIC_FEATURES].to_numpy(dtype=np.float32, na_value=np.nan)
    return codes, np.ascontiguousarray(nums)
