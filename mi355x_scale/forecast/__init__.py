"""W1: demand forecasting — SARIMAX-lite (CPU reference), per-SKU tuning
pipeline, and the batched CDNA4 GPU fit."""

from .sarimax import SARIMAX, SARIMAXResults  # noqa: F401
from .pipeline import (add_exo_variables, split_train_score_data,  # noqa: F401
                       evaluate_model, build_tune_and_score_model,
                       run_fine_grained_forecast,
                       run_fine_grained_forecast_gpu,
                       run_fine_grained_forecast_sharded,
                       read_forecast_shards, TUNING_SCHEMA,
                       FORECAST_HORIZON, SEARCH_SPACE)
from .holtwinters import ExponentialSmoothing, HoltWintersResults  # noqa: F401
