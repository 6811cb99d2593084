"""Small shared utilities: timing spans and percentile math."""

from kakveda_amd.utils.timing import Stopwatch, percentiles  # noqa: F401
