"""Named, *recognizable* function objects used as DSL defaults.

The reference's API is lambda-everywhere (e.g. ``count(key=lambda x: x)``,
reference: dampr.py:439-448); opaque lambdas can only run on the host.  The
MI355X engine lowers a stage to device kernels when the functions it was
built from are these named objects (or a handful of stdlib ones:
``operator.add``, builtin ``min``/``max``) — identity-preserving `is`
checks, never bytecode inspection.  Users who pass their own callables get
the host path for that stage automatically; passing these (or just using
the DSL defaults) gets the device path.
"""
import operator


def identity(x):
    return x


def one(_x):
    return 1


def fst(x):
    return x[0]


def snd(x):
    return x[1]


add = operator.add
mul = operator.mul


# --- recognition tables ----------------------------------------------------

# binop -> segmented-reduce op name understood by the device backend
ASSOC_BINOPS = {
    add: "sum",
    operator.add: "sum",
    min: "min",
    max: "max",
}

# key/value extractors the columnar engine can apply without host trips
# (fst/snd await pair-column support in the engine)
COLUMN_FUNCS = {
    identity: "identity",
    one: "one",
}


_TOKEN_RE = None


def tokenize_set(line):
    """Distinct ASCII word-tokens of a line, lowercased — the host mirror
    of the device tokenizer (ops/hip: is_word/lower_ascii).  Built-in so
    ``device_text(...).flat_map(funcs.tokenize_set).count()`` lowers to
    the fused single-pass document-frequency kernel."""
    global _TOKEN_RE
    if _TOKEN_RE is None:
        import re
        _TOKEN_RE = re.compile(r"[a-z0-9_]+")
    return set(_TOKEN_RE.findall(line.lower()))


# --- join pair aggregates ---------------------------------------------------
# PJoin.reduce(aggregate, many=True) aggregates the cartesian product of
# each key's (left values, right values).  These named per-pair forms let
# the device engine emit the product straight from the hash-join kernel.
#
# The right side is materialized first: the engines hand one-pass group
# iterators to aggregates (as the reference does), so a bare nested
# comprehension would exhaust `right` after the first left value and
# silently drop pairs.  These named funcs are defined as the FULL cross
# product — identical on the host path and the hash-join kernel.

def pair_sum(left, right):
    right = list(right)
    return [lv + rv for lv in left for rv in right]


def pair_product(left, right):
    right = list(right)
    return [lv * rv for lv in left for rv in right]


def pair_left(left, right):
    right = list(right)
    return [lv for lv in left for _rv in right]


def pair_right(left, right):
    right = list(right)
    return [rv for _lv in left for rv in right]


JOIN_PAIR_FUNCS = {
    pair_sum: "sum",
    pair_product: "mul",
    pair_left: "left",
    pair_right: "right",
}


# cross-join (K9) apply ops the device can fuse over the broadcast
# product; all COMMUTATIVE, so cross_left/cross_right share one table
# (cross_right wraps its lambda with swapped operands).
CROSS_BINOPS = {
    add: "add",
    operator.add: "add",
    mul: "mul",
    operator.mul: "mul",
    min: "min",
    max: "max",
}


# cross_set aggregate: the whole broadcast side folds to ONE scalar
SET_AGGS = {
    sum: "sum",
    min: "min",
    max: "max",
}


def set_agg_name(f):
    try:
        return SET_AGGS.get(f)
    except TypeError:
        return None


def cross_binop_name(f):
    try:
        return CROSS_BINOPS.get(f)
    except TypeError:
        return None


def join_pair_name(f):
    try:
        return JOIN_PAIR_FUNCS.get(f)
    except TypeError:
        return None


def binop_name(f):
    """Device op name for an associative binop, or None."""
    try:
        return ASSOC_BINOPS.get(f)
    except TypeError:
        return None


def column_func_name(f):
    try:
        return COLUMN_FUNCS.get(f)
    except TypeError:
        return None
