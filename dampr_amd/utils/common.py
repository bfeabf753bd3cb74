"""Common pipeline idioms (role parity: reference dampr/utils/common.py)."""


def filter_by_count(pipe, key_func, filter_func):
    """Keep records whose key's multiplicity is accepted by
    ``filter_func(count)``.

    Shape: group the records under their key once, compute the accepted
    key set from a count pass, then an equi-join keeps exactly the
    groups whose key survived — the record side rides the LEFT of the
    join and its group iterator is emitted directly.
    """
    grouped = pipe.group_by(key_func)
    accepted = pipe.map(key_func) \
        .count() \
        .filter(lambda kc: filter_func(kc[1])) \
        .group_by(lambda kc: kc[0], lambda kc: kc[1])
    return grouped.join(accepted) \
        .reduce(lambda values, _counts: values, many=True) \
        .map(lambda kv: kv[1])
