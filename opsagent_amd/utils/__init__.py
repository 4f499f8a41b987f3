from opsagent_amd.utils.perf import PerfStats, get_perf_stats
from opsagent_amd.utils.jsonrepair import clean_json, extract_field, parse_json
from opsagent_amd.utils.yamlextract import extract_yaml

__all__ = [
    "PerfStats",
    "get_perf_stats",
    "clean_json",
    "extract_field",
    "parse_json",
    "extract_yaml",
]
