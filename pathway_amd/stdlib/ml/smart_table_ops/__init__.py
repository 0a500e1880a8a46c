"""Fuzzy join (reference stdlib/ml/smart_table_ops/_fuzzy_join.py).

The reference algorithm, reproduced as real dataflow (incremental under
updates, not a host recompute):

1. feature generation per row (tokenize / letters) -> edge table
   (node, feature, weight);
2. per-feature frequency -> normalization weight (WEIGHT 1/2^ceil(log2 n),
   LOGWEIGHT 1/ceil(log2(n+1)), NONE n);
3. heavy/light split at HEAVY_LIGHT_THRESHOLD: light features join
   directly, heavy features only score candidate pairs the light pass
   already produced (the reference's candidate pruning);
4. pair weight = sum over shared features of w_l * w_r * feature_weight;
5. mutual-best matching: argmax per left then per right (deviation:
   ties break by row id instead of the reference's pseudoweight tuple);
6. by_hand_match pre-filters matched nodes and overrides final rows.

Deviation from the reference's storage: edge/feature tables are keyed by
the feature VALUE (joined on it) rather than by feature pointer — same
semantics, fewer pointer indirections.
"""
from __future__ import annotations

import enum
import math
from typing import Any, Callable

import pathway_amd.internals.common as common
from pathway_amd import reducers
from pathway_amd.internals import thisclass
from pathway_amd.internals.table import Table

this = thisclass.this
left = thisclass.left
right = thisclass.right


class JoinType(enum.Enum):
    FULL = 0
    LEFT = 1
    RIGHT = 2


def _tokenize(obj: Any) -> Any:
    return str(obj).split()


def _letters(obj: Any) -> Any:
    return [c.lower() for c in str(obj) if c.isalnum()]


class FuzzyJoinFeatureGeneration(enum.IntEnum):
    AUTO = enum.auto()
    TOKENIZE = enum.auto()
    LETTERS = enum.auto()

    @property
    def generate(self) -> Callable[[Any], Any]:
        if self == FuzzyJoinFeatureGeneration.LETTERS:
            return _letters
        return _tokenize


def _discrete_weight(cnt: float) -> float:
    return 0.0 if cnt == 0 else 1 / (2 ** math.ceil(math.log2(cnt)))


def _discrete_logweight(cnt: float) -> float:
    return 0.0 if cnt == 0 else 1 / math.ceil(math.log2(cnt + 1))


class FuzzyJoinNormalization(enum.IntEnum):
    WEIGHT = enum.auto()
    LOGWEIGHT = enum.auto()
    NONE = enum.auto()

    @property
    def normalize(self) -> Callable[[Any], Any]:
        if self == FuzzyJoinNormalization.WEIGHT:
            return _discrete_weight
        if self == FuzzyJoinNormalization.LOGWEIGHT:
            return _discrete_logweight
        return lambda cnt: cnt


def _normalize_weight(cnt: float, normalization_type: int) -> float:
    return float(FuzzyJoinNormalization(normalization_type).normalize(cnt))


def _ptr_lt(a, b) -> bool:
    return repr(a) < repr(b)


def _ptr_ne(a, b) -> bool:
    return repr(a) != repr(b)


def _edges_for(tab: Table, col, feature_generation) -> Table:
    e = tab.select(feature=common.apply(feature_generation.generate, col))
    e = e.flatten(this.feature, origin_id="origin_id")
    return e.select(node=this.origin_id, feature=this.feature, weight=1.0)


def smart_fuzzy_match(
    left_col,
    right_col,
    *,
    by_hand_match: Table | None = None,
    normalization=FuzzyJoinNormalization.LOGWEIGHT,
    feature_generation=FuzzyJoinFeatureGeneration.AUTO,
    HEAVY_LIGHT_THRESHOLD: int = 100,
) -> Table:
    """Best-pair fuzzy match between two columns (reference
    _fuzzy_join.py:199-246)."""
    ltab, rtab = left_col.table, right_col.table
    self_match = ltab is rtab and left_col.name == right_col.name
    tabs = [ltab] if self_match else [ltab, rtab]
    cols = [left_col] if self_match else [left_col, right_col]
    processed = []
    features: Table | None = None
    for tab, col in zip(tabs, cols):
        edges = _edges_for(tab, col, feature_generation)
        feats = edges.groupby(this.feature).reduce(
            feature=this.feature,
            normalization_type=int(normalization),
            weight=1.0,
        )
        features = feats if features is None else features.update_rows(feats)
        processed.append(edges)
    assert features is not None
    if self_match:
        return fuzzy_self_match(
            processed[0], features, by_hand_match, HEAVY_LIGHT_THRESHOLD
        )
    return fuzzy_match(
        processed[0], processed[1], features, by_hand_match,
        HEAVY_LIGHT_THRESHOLD,
    )


def fuzzy_self_match(
    edges: Table, features: Table, by_hand_match: Table | None = None,
    HEAVY_LIGHT_THRESHOLD: int = 100,
) -> Table:
    return _fuzzy_match(
        edges, edges, features, symmetric=True,
        HEAVY_LIGHT_THRESHOLD=HEAVY_LIGHT_THRESHOLD,
        by_hand_match=by_hand_match,
    )


def fuzzy_match(
    edges_left: Table, edges_right: Table, features: Table,
    by_hand_match: Table | None = None, HEAVY_LIGHT_THRESHOLD: int = 100,
) -> Table:
    return _fuzzy_match(
        edges_left, edges_right, features, symmetric=False,
        HEAVY_LIGHT_THRESHOLD=HEAVY_LIGHT_THRESHOLD,
        by_hand_match=by_hand_match,
    )


def fuzzy_match_with_hint(
    edges_left: Table, edges_right: Table, features: Table,
    by_hand_match: Table, HEAVY_LIGHT_THRESHOLD: int = 100,
) -> Table:
    return _fuzzy_match(
        edges_left, edges_right, features, symmetric=False,
        HEAVY_LIGHT_THRESHOLD=HEAVY_LIGHT_THRESHOLD,
        by_hand_match=by_hand_match,
    )


def _filter_unmatched(edges: Table, matched_nodes: Table) -> Table:
    """Drop edges whose node was already matched by hand (reference
    _filter_out_matched_by_hand)."""
    j = edges.join_left(
        matched_nodes, edges.node == matched_nodes.node
    ).select(
        node=edges.node,
        feature=edges.feature,
        weight=edges.weight,
        hit=matched_nodes.node,
    )
    return j.filter(this.hit.is_none()).select(
        this.node, this.feature, this.weight
    )


def _fuzzy_match(
    edges_left: Table,
    edges_right: Table,
    features: Table,
    symmetric: bool,
    HEAVY_LIGHT_THRESHOLD: int,
    by_hand_match: Table | None = None,
) -> Table:
    thr = HEAVY_LIGHT_THRESHOLD
    if by_hand_match is not None:
        ml = by_hand_match.select(node=this.left)
        mr = by_hand_match.select(node=this.right)
        edges_left = _filter_unmatched(edges_left, ml)
        edges_right = (
            edges_left if symmetric else _filter_unmatched(edges_right, mr)
        )

    edges_all = (
        edges_left if symmetric
        else Table.concat_reindex(edges_left, edges_right)
    )
    cnts = edges_all.groupby(this.feature).reduce(
        feature=this.feature, cnt=reducers.count()
    )
    feats = features.join(cnts, features.feature == cnts.feature).select(
        feature=features.feature,
        fweight=features.weight
        * common.apply(_normalize_weight, cnts.cnt, features.normalization_type),
        cnt=cnts.cnt,
    )

    def annotate(edges: Table) -> Table:
        return edges.join(feats, edges.feature == feats.feature).select(
            node=edges.node,
            feature=edges.feature,
            weight=edges.weight,
            fweight=feats.fweight,
            cnt=feats.cnt,
        )

    el = annotate(edges_left)
    er = el if symmetric else annotate(edges_right)
    light_l, heavy_l = el.filter(this.cnt < thr), el.filter(this.cnt >= thr)
    light_r, heavy_r = (
        (light_l.copy(), heavy_l.copy()) if symmetric
        else (er.filter(this.cnt < thr), er.filter(this.cnt >= thr))
    )

    # light x light: direct join on the feature
    ll = light_l.join(light_r, light_l.feature == light_r.feature).select(
        left=light_l.node,
        right=light_r.node,
        weight=light_l.weight * light_r.weight * light_l.fweight,
    )
    if symmetric:
        ll = ll.filter(common.apply(_ptr_ne, this.left, this.right))
    light_pairs = ll.groupby(this.left, this.right).reduce(
        this.left, this.right, weight=reducers.sum(this.weight)
    )

    # heavy features: only score pairs the light pass already produced
    hv = (
        light_pairs.join(heavy_l, light_pairs.left == heavy_l.node)
        .select(this.left, this.right, hfeature=heavy_l.feature,
                lw=heavy_l.weight, lfw=heavy_l.fweight)
        .join(
            heavy_r,
            this.right == heavy_r.node,
            this.hfeature == heavy_r.feature,
        )
        .select(
            this.left, this.right,
            weight=this.lw * heavy_r.weight * this.lfw,
        )
    )
    node_node = Table.concat_reindex(light_pairs, hv).groupby(
        this.left, this.right
    ).reduce(this.left, this.right, weight=reducers.sum(this.weight))

    # mutual best: argmax per left, then per right
    bl = node_node.groupby(this.left).reduce(
        left=this.left,
        best=reducers.argmax(this.weight),
        weight=reducers.max(this.weight),
    )
    bl = bl.select(this.left, this.weight, right=node_node.ix(bl.best).right)
    br = bl.groupby(this.right).reduce(
        right=this.right,
        best=reducers.argmax(this.weight),
        weight=reducers.max(this.weight),
    )
    out = br.select(this.right, this.weight, left=bl.ix(br.best).left)
    out = out.select(this.left, this.right, this.weight)
    if symmetric:
        out = out.filter(common.apply(_ptr_lt, this.left, this.right))
    if by_hand_match is not None:
        out = out.update_rows(by_hand_match)
    return out


def _concat_desc(table: Table) -> Table:
    def concat_columns(*args) -> str:
        return " ".join(str(a) for a in args)

    return table.select(
        desc=common.apply(
            concat_columns, *[table[n] for n in table.column_names()]
        )
    )


def _fuzzy_match_tables(
    left_table: Table,
    right_table: Table,
    *,
    by_hand_match: Table | None = None,
    normalization=FuzzyJoinNormalization.LOGWEIGHT,
    feature_generation=FuzzyJoinFeatureGeneration.AUTO,
) -> Table:
    lt = _concat_desc(left_table)
    rt = _concat_desc(right_table)
    return smart_fuzzy_match(
        lt.desc, rt.desc,
        by_hand_match=by_hand_match,
        normalization=normalization,
        feature_generation=feature_generation,
    )


def fuzzy_match_tables(
    left_table: Table,
    right_table: Table,
    *,
    by_hand_match: Table | None = None,
    normalization=FuzzyJoinNormalization.LOGWEIGHT,
    feature_generation=FuzzyJoinFeatureGeneration.AUTO,
    left_projection: dict[str, str] | None = None,
    right_projection: dict[str, str] | None = None,
) -> Table:
    """Match rows of two tables by fuzzy column content (reference
    _fuzzy_join.py:106-176).  With projections, each bucket of columns is
    matched independently and pair weights are summed."""
    left_projection = left_projection or {}
    right_projection = right_projection or {}
    if not left_projection or not right_projection:
        return _fuzzy_match_tables(
            left_table=left_table, right_table=right_table,
            by_hand_match=by_hand_match, normalization=normalization,
            feature_generation=feature_generation,
        )
    buckets: dict[str, tuple[list, list]] = {}
    for col, b in left_projection.items():
        buckets.setdefault(b, ([], []))[0].append(col)
    for col, b in right_projection.items():
        buckets.setdefault(b, ([], []))[1].append(col)
    parts = []
    for b, (lcols, rcols) in buckets.items():
        if not lcols or not rcols:
            continue
        parts.append(
            _fuzzy_match_tables(
                left_table=left_table[lcols],
                right_table=right_table[rcols],
                by_hand_match=by_hand_match,
                normalization=normalization,
                feature_generation=feature_generation,
            )
        )
    matchings = Table.concat_reindex(*parts)
    return matchings.groupby(matchings.left, matchings.right).reduce(
        matchings.left, matchings.right,
        weight=reducers.sum(matchings.weight),
    )


# round-1 compatibility alias (token-overlap greedy matcher is superseded
# by the reference algorithm above)
smart_fuzzy_join = smart_fuzzy_match
