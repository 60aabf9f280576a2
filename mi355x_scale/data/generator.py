"""Synthetic dataset generators.

Re-creates, natively, what the reference's data-gen notebooks produce:

* ``generate_demand_data`` — per-SKU weekly ARMA demand series with a
  pre-COVID sqrt trend, COVID decay, and Christmas / New-Year
  multipliers (reference: ``group_apply/_resources/01-data-generator.py``
  :131-183 factor frame, :197-226 per-product ARMA params,
  :242-254 ``generate_arma``, :276-306 per-SKU series + factors).
  Seeded with 123 like the reference so results are comparable.
* ``generate_bom`` — random 3-level bill-of-materials DAG per SKU
  (reference: ``01-data-generator.py:361-543``).
* ``write_image_parquet`` — synthetic "ImageNet-shaped" parquet for the
  W3 streaming benchmark (uint8 HWC images + int label), replacing the
  Kaggle/Delta ingest of ``deep_learning/1.data-preparation.py`` (no
  network here; BASELINE.json mandates synthetic data).
"""
from __future__ import annotations

import os
import string
from typing import Dict, List, Optional, Tuple

import numpy as np
import pandas as pd
import pyarrow as pa
import pyarrow.parquet as pq
from scipy.signal import lfilter

SEED = 123  # the reference's seed (01-data-generator.py:101,207,243,385)


# --------------------------------------------------------------------------
# W1: demand series
# --------------------------------------------------------------------------
def arma_generate_sample(ar, ma, nsample: int, scale: float = 1.0,
                         burnin: int = 3000,
                         rng: Optional[np.random.Generator] = None) -> np.ndarray:
    """ARMA(p,q) sample path: x = lfilter(ma, ar, scale*eps), burn-in dropped.

    Same convention as statsmodels ``arma_generate_sample`` (ar/ma include
    the leading 1; ar lags enter negated), which the reference calls at
    ``01-data-generator.py:246`` with burnin 3000.
    """
    rng = rng or np.random.default_rng(SEED)
    eps = scale * rng.standard_normal(nsample + burnin)
    x = lfilter(np.asarray(ma, dtype=np.float64),
                np.asarray(ar, dtype=np.float64), eps)
    return x[burnin:]


def arma_generate_sample_batch_gpu(ar, ma, nsample: int,
                                   scale=1.0, burnin: int = 3000,
                                   rng: Optional[np.random.Generator] = None,
                                   device="cuda:0"):
    """Batched GPU form of :func:`arma_generate_sample`:
    one group per lane, time-major [T][G] output.

    ``ar``/``ma``: [G, na]/[G, nb] float arrays with the statsmodels
    leading-1 convention; ``scale``: scalar or [G]. Returns a torch
    tensor [nsample, G] on ``device``.
    """
    import torch

    from ..ops import _C, require_ext
    require_ext()
    ar_t = torch.as_tensor(np.asarray(ar, dtype=np.float32),
                           device=device).contiguous()
    ma_t = torch.as_tensor(np.asarray(ma, dtype=np.float32),
                           device=device).contiguous()
    G = ar_t.shape[0]
    rng = rng or np.random.default_rng(SEED)
    eps = rng.standard_normal((nsample + burnin, G)).astype(np.float32)
    eps = eps * np.broadcast_to(np.asarray(scale, dtype=np.float32),
                                (G,))[None, :]
    eps_t = torch.as_tensor(np.ascontiguousarray(eps), device=device)
    out = torch.empty((nsample, G), dtype=torch.float32, device=device)
    _C.arma_generate(ar_t, ma_t, eps_t, out, burnin)
    return out


def _week_dates(n_weeks: int, end: str = "2023-06-26") -> pd.DatetimeIndex:
    endts = pd.Timestamp(end)
    return pd.date_range(end=endts, periods=n_weeks, freq="W-MON")


def demand_factors(dates: pd.DatetimeIndex,
                   covid_start: str = "2020-03-01",
                   covid_len_weeks: int = 16) -> pd.DataFrame:
    """Shared date/factor frame (reference ``01-data-generator.py:131-183``):
    covid decay window, christmas and new-year indicator weeks."""
    df = pd.DataFrame({"Date": dates})
    cs = pd.Timestamp(covid_start)
    weeks_since = ((df["Date"] - cs).dt.days // 7).clip(lower=-1)
    in_covid = (weeks_since >= 0) & (weeks_since < covid_len_weeks)
    df["covid"] = np.where(in_covid, np.exp(-0.25 * weeks_since.clip(lower=0)), 0.0)
    df["christmas"] = ((df["Date"].dt.month == 12) & (df["Date"].dt.day >= 18)).astype(float)
    df["new_year"] = ((df["Date"].dt.month == 1) & (df["Date"].dt.day <= 7)).astype(float)
    return df


def generate_demand_data(n_products: int = 5, skus_per_product: int = 10,
                         n_weeks: int = 157, seed: int = SEED) -> pd.DataFrame:
    """Weekly demand for n_products × skus_per_product SKUs.

    Columns: Product, SKU, Date, Demand, covid, christmas, new_year —
    the schema W1's ``applyInPandas`` pipeline consumes
    (``group_apply/02_Fine_Grained_Demand_Forecasting.py:343-370``).
    """
    rng = np.random.default_rng(seed)
    dates = _week_dates(n_weeks)
    factors = demand_factors(dates)
    rows: List[pd.DataFrame] = []
    for p in range(n_products):
        # Per-product ARMA parameters (stable region), ref :197-226.
        ar1 = rng.uniform(0.3, 0.8)
        ma1 = rng.uniform(-0.4, 0.4)
        base = rng.uniform(400, 1200)
        trend_amp = rng.uniform(2.0, 8.0)
        product = f"P{p:03d}"
        for s in range(skus_per_product):
            sku = f"{product}_SKU{s:04d}"
            noise = arma_generate_sample([1.0, -ar1], [1.0, ma1], n_weeks,
                                         scale=base * 0.05, rng=rng)
            t = np.arange(n_weeks, dtype=np.float64)
            level = base + trend_amp * np.sqrt(t) + noise
            # COVID drop + seasonal multipliers (ref :295-306).
            level = level * (1.0 - 0.5 * factors["covid"].to_numpy())
            level = level * (1.0 + 0.35 * factors["christmas"].to_numpy())
            level = level * (1.0 - 0.20 * factors["new_year"].to_numpy())
            demand = np.clip(np.round(level), 0, None)
            df = factors.copy()
            df.insert(0, "SKU", sku)
            df.insert(0, "Product", product)
            df["Demand"] = demand
            rows.append(df)
    out = pd.concat(rows, ignore_index=True)
    return out[["Product", "SKU", "Date", "Demand", "covid", "christmas", "new_year"]]


# --------------------------------------------------------------------------
# W1: bill-of-materials DAG (reference 01-data-generator.py:361-543)
# --------------------------------------------------------------------------
def _rand_ids(rng: np.random.Generator, n: int, length: int = 8) -> List[str]:
    alphabet = np.array(list(string.ascii_uppercase + string.digits))
    return ["".join(rng.choice(alphabet, size=length)) for _ in range(n)]


def generate_bom(skus: List[str], levels: int = 3,
                 children: Tuple[int, int] = (2, 4),
                 seed: int = SEED) -> Tuple[pd.DataFrame, pd.DataFrame]:
    """Random BoM DAG per SKU: ``levels`` deep, 2–4 children per node.

    Returns (bom, sku_mapper): bom has material_in/material_out/qty edges;
    sku_mapper maps each SKU to its root material id.
    """
    import networkx as nx

    rng = np.random.default_rng(seed)
    edges = []
    mapper = []
    for sku in skus:
        root = f"M_{sku}"
        mapper.append({"sku": sku, "final_mat_number": root})
        frontier = [root]
        sku_edges = []
        for _ in range(levels):
            nxt = []
            for node in frontier:
                k = int(rng.integers(children[0], children[1] + 1))
                for child in _rand_ids(rng, k):
                    qty = int(rng.integers(1, 5))
                    sku_edges.append({"material_in": child,
                                      "material_out": node, "qty": qty})
                    nxt.append(child)
            frontier = nxt
        # Validate THIS SKU's subgraph only (O(edges-per-SKU), not the
        # cumulative edge list of every SKU generated so far).
        g = nx.DiGraph()
        g.add_edges_from((e["material_in"], e["material_out"])
                         for e in sku_edges)
        assert nx.is_directed_acyclic_graph(g)
        edges.extend(sku_edges)
    return pd.DataFrame(edges), pd.DataFrame(mapper)


# --------------------------------------------------------------------------
# W3: synthetic image parquet
# --------------------------------------------------------------------------
def write_image_parquet(
    out_dir: str,
    num_rows: int = 2048,
    image_hw: Tuple[int, int] = (224, 224),
    num_classes: int = 1000,
    rows_per_group: int = 64,
    rows_per_file: int = 512,
    seed: int = SEED,
) -> str:
    """Write a synthetic image dataset: uint8 HWC images (flattened bytes
    column) + int64 label. Schema mirrors what the reference's Delta table
    feeds Petastorm (binary image content + object_id label,
    ``deep_learning/1.data-preparation.py:118-205``)."""
    os.makedirs(out_dir, exist_ok=True)
    h, w = image_hw
    row_bytes = h * w * 3
    rng = np.random.default_rng(seed)
    file_idx = 0
    written = 0
    while written < num_rows:
        n = min(rows_per_file, num_rows - written)
        # One random block, sliced per row (cheap to generate, incompressible
        # like JPEG payloads so decode cost is realistic).
        blob = rng.integers(0, 256, size=(n, row_bytes), dtype=np.uint8)
        labels = rng.integers(0, num_classes, size=n, dtype=np.int64)
        table = pa.table({
            "image": pa.array([r.tobytes() for r in blob],
                              type=pa.binary(row_bytes)),
            "label": pa.array(labels),
        })
        path = os.path.join(out_dir, f"part-{file_idx:05d}.parquet")
        pq.write_table(table, path, row_group_size=rows_per_group,
                       compression="none")
        file_idx += 1
        written += n
    return out_dir


def write_jpeg_parquet(
    out_dir: str,
    num_rows: int = 2048,
    source_hw_range: Tuple[int, int] = (256, 384),
    num_classes: int = 1000,
    rows_per_group: int = 64,
    rows_per_file: int = 512,
    quality: int = 85,
    seed: int = SEED,
) -> str:
    """Write a synthetic dataset of ENCODED JPEG bytes (variable-size
    binary column) + label — the reference's actual Delta schema (raw
    ``content`` bytes from ``binaryFile``,
    ``deep_learning/1.data-preparation.py:118-124``), for exercising the
    loader under the real per-row decode cost the reference's
    TransformSpec pays every epoch (``deep_learning/2...py:282-296``).
    Images are smooth gradients + noise at varying source sizes so JPEG
    encode/decode cost and the resize path are realistic.
    """
    import io

    from PIL import Image

    os.makedirs(out_dir, exist_ok=True)
    rng = np.random.default_rng(seed)
    file_idx = 0
    written = 0
    lo, hi = source_hw_range
    while written < num_rows:
        n = min(rows_per_file, num_rows - written)
        blobs = []
        for _ in range(n):
            h = int(rng.integers(lo, hi + 1))
            w = int(rng.integers(lo, hi + 1))
            yy, xx = np.mgrid[0:h, 0:w]
            base = np.stack([
                (127 + 120 * np.sin(xx / rng.uniform(8, 40))),
                (127 + 120 * np.cos(yy / rng.uniform(8, 40))),
                (127 + 120 * np.sin((xx + yy) / rng.uniform(8, 40))),
            ], axis=-1)
            img = np.clip(base + rng.normal(0, 12, size=(h, w, 3)),
                          0, 255).astype(np.uint8)
            buf = io.BytesIO()
            Image.fromarray(img).save(buf, "JPEG", quality=quality)
            blobs.append(buf.getvalue())
        labels = rng.integers(0, num_classes, size=n, dtype=np.int64)
        table = pa.table({
            "image": pa.array(blobs, type=pa.binary()),
            "label": pa.array(labels),
        })
        path = os.path.join(out_dir, f"part-{file_idx:05d}.parquet")
        pq.write_table(table, path, row_group_size=rows_per_group,
                       compression="none")
        file_idx += 1
        written += n
    return out_dir


def decode_jpeg_batch(pdf: pd.DataFrame,
                      image_hw: Tuple[int, int] = (224, 224)
                      ) -> Dict[str, np.ndarray]:
    """Reference-style per-row CPU transform (``deep_learning/2...py:
    282-296``): JPEG decode → resize(short side 256-scaled) →
    center-crop. Output stays uint8 NHWC — normalization/CHW runs on
    device in the fused HIP kernel, where the reference burned CPU."""
    import io

    from PIL import Image

    h, w = image_hw
    out = np.empty((len(pdf), h, w, 3), dtype=np.uint8)
    for i, b in enumerate(pdf["image"]):
        with Image.open(io.BytesIO(b)) as im:
            im = im.convert("RGB")
            scale = max(h / im.height, w / im.width) * 256.0 / 224.0
            nh = int(round(im.height * scale))
            nw = int(round(im.width * scale))
            im = im.resize((nw, nh))
            left = (nw - w) // 2
            top = (nh - h) // 2
            out[i] = np.asarray(im.crop((left, top, left + w, top + h)),
                                dtype=np.uint8)
    return {"image": out, "label": pdf["label"].to_numpy()}


def decode_image_batch(pdf: pd.DataFrame, image_hw: Tuple[int, int] = (224, 224)) -> Dict[str, np.ndarray]:
    """Default transform for the synthetic image dataset: bytes → uint8
    NHWC array. Normalization/CHW happens on-device (fused HIP kernel) —
    unlike the reference, which burns CPU in PIL/torchvision transforms
    (``deep_learning/2...py:282-296``)."""
    h, w = image_hw
    imgs = np.stack([
        np.frombuffer(b, dtype=np.uint8).reshape(h, w, 3)
        for b in pdf["image"]
    ])
    return {"image": imgs, "label": pdf["label"].to_numpy()}
