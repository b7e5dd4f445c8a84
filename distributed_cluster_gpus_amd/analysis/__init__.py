from .aggregate import load_run, aggregate_cluster, summarize_run
from .plots import comparison_report, COMPARISON_FIGURES
from .plots_single import single_algo_report, SINGLE_FIGURES
from .montecarlo import population_report, population_frame

__all__ = ["load_run", "aggregate_cluster", "summarize_run",
           "comparison_report", "COMPARISON_FIGURES",
           "single_algo_report", "SINGLE_FIGURES",
           "population_report", "population_frame"]
