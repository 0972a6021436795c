from . import metrics, stats, trace  # noqa
