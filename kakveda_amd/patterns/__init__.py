from kakveda_amd.patterns.kmeans import StreamingKMeans  # noqa: F401
from kakveda_amd.patterns.miner import PatternMiner  # noqa: F401
