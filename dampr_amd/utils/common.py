"""Common pipeline idioms (role parity: reference dampr/utils/common.py)."""


def filter_by_count(pipe, key_func, filter_func):
    """Keep records whose key appears a number of times accepted by
    ``filter_func(count)`` — the count → join-back idiom."""
    item_count = pipe.map(key_func) \
                     .count() \
                     .filter(lambda count: filter_func(count[1]))

    return item_count.group_by(lambda x: x[0], lambda x: x[1]) \
                     .join(pipe.group_by(key_func)) \
                     .reduce(lambda lit, rit: rit, many=True) \
                     .map(lambda x: x[1])
