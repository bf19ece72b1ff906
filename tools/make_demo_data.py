#!/usr/bin/env python3
"""Generate the synthetic sales + segmentation demo datasets used by
config/configs_sales_supervised.yaml and
config/configs_segmentation_unsupervised.yaml.

The reference ships real datasets under examples/data/{sales_dataset,
segmentation_dataset} (same schemas); no datasets are fetchable in this
stack, so these generators produce schema-compatible synthetic data:
csv/ (main), source/ (drift source snapshot), stability_index/0..8
(9 snapshots) and data_dictionary.csv.
"""

from __future__ import annotations

import argparse
import os

import numpy as np
import pandas as pd


def _write(pdf: pd.DataFrame, d: str):
    os.makedirs(d, exist_ok=True)
    pdf.to_csv(os.path.join(d, "part-00000.csv"), index=False)


def make_sales(root: str, rows: int, seed: int = 7):
    rng = np.random.default_rng(seed)

    def frame(n, shift=0.0):
        item_type = rng.choice(["Dairy", "Soft Drinks", "Meat", "Fruits", "Household", "Baking Goods"], n)
        weight = np.round(rng.normal(12 + shift, 4, n).clip(2, 25), 3)
        weight[rng.random(n) < 0.05] = np.nan
        mrp = np.round(rng.normal(140 + 10 * shift, 60, n).clip(30, 270), 4)
        vis = np.round(rng.beta(2, 20, n), 6)
        years = rng.integers(1, 30, n)
        sales_num = mrp * (1 + 0.1 * years) * rng.lognormal(0, 0.3, n)
        return pd.DataFrame(
            {
                "Item_Identifier": [f"FD{i:05d}" for i in rng.integers(0, max(n // 2, 1), n)],
                "Item_Weight": weight,
                "Item_Fat_Content": rng.choice(["Low Fat", "Regular", "low fat", "LF"], n, p=[0.5, 0.35, 0.1, 0.05]),
                "Item_Visibility": vis,
                "Item_Type": item_type,
                "Item_MRP": mrp,
                "Outlet_Identifier": rng.choice([f"OUT{i:03d}" for i in range(10)], n),
                "Outlet_Size": rng.choice(["Small", "Medium", "High"], n, p=[0.4, 0.45, 0.15]),
                "Outlet_Location_Type": rng.choice(["Tier 1", "Tier 2", "Tier 3"], n),
                "Outlet_Type": rng.choice(["Grocery Store", "Supermarket Type1", "Supermarket Type2"], n),
                "Item_Category": rng.choice(["Food", "Drinks", "Non-Consumable"], n, p=[0.6, 0.2, 0.2]),
                "Outlet_Years": years,
                "sales": np.where(sales_num > np.median(sales_num), ">2k", "<=2k"),
            }
        )

    _write(frame(rows), os.path.join(root, "csv"))
    _write(frame(rows, shift=1.5), os.path.join(root, "source"))
    for i in range(9):
        _write(frame(max(rows // 2, 200), shift=0.2 * i), os.path.join(root, "stability_index", str(i)))
    dd = pd.DataFrame(
        {
            "Attribute": ["Item_Identifier", "Item_Weight", "Item_Fat_Content", "Item_Visibility", "Item_Type",
                          "Item_MRP", "Outlet_Identifier", "Outlet_Size", "Outlet_Location_Type", "Outlet_Type",
                          "Item_Category", "Outlet_Years", "sales"],
            "Description": ["item id", "weight of item", "fat content", "display visibility share", "item type",
                            "max retail price", "outlet id", "outlet size", "outlet location tier", "outlet type",
                            "item category", "years outlet active", "sales above 2k indicator"],
        }
    )
    os.makedirs(root, exist_ok=True)
    dd.to_csv(os.path.join(root, "data_dictionary.csv"), index=False)


def make_segmentation(root: str, rows: int, seed: int = 11):
    rng = np.random.default_rng(seed)

    def frame(n, shift=0.0):
        age = rng.integers(18, 90, n).astype(float)
        age[rng.random(n) < 0.03] = np.nan
        wexp = np.round(rng.gamma(2, 2 + shift, n), 1)
        wexp[rng.random(n) < 0.1] = np.nan
        fam = rng.integers(1, 9, n).astype(float)
        fam[rng.random(n) < 0.05] = np.nan
        return pd.DataFrame(
            {
                "ID": rng.integers(100000, 999999, n),
                "Gender": rng.choice(["Male", "Female"], n),
                "Ever_Married": rng.choice(["Yes", "No"], n, p=[0.6, 0.4]),
                "Age": age,
                "Graduated": rng.choice(["Yes", "No"], n),
                "Profession": rng.choice(["Healthcare", "Engineer", "Lawyer", "Artist", "Doctor", "Executive"], n),
                "Work_Experience": wexp,
                "Spending_Score": rng.choice(["Low", "Average", "High"], n, p=[0.5, 0.3, 0.2]),
                "Family_Size": fam,
                "Var_1": rng.choice([f"Cat_{i}" for i in range(1, 8)], n),
            }
        )

    _write(frame(rows), os.path.join(root, "csv"))
    _write(frame(rows, shift=0.8), os.path.join(root, "source"))
    for i in range(9):
        _write(frame(max(rows // 2, 200), shift=0.1 * i), os.path.join(root, "stability_index", str(i)))
    dd = pd.DataFrame(
        {
            "Attribute": ["ID", "Gender", "Ever_Married", "Age", "Graduated", "Profession",
                          "Work_Experience", "Spending_Score", "Family_Size", "Var_1"],
            "Description": ["customer id", "gender", "ever married", "age in years", "graduated",
                            "profession", "years of work experience", "spending score band",
                            "family size", "anonymous category"],
        }
    )
    os.makedirs(root, exist_ok=True)
    dd.to_csv(os.path.join(root, "data_dictionary.csv"), index=False)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="data")
    ap.add_argument("--rows", type=int, default=3000)
    ap.add_argument("--which", default="both", choices=["sales", "segmentation", "both"])
    a = ap.parse_args()
    if a.which in ("sales", "both"):
        make_sales(os.path.join(a.out, "sales_dataset"), a.rows)
    if a.which in ("segmentation", "both"):
        make_segmentation(os.path.join(a.out, "segmentation_dataset"), a.rows)
    # shared metric dictionary used by report_generation
    md = os.path.join(a.out, "metric_dictionary.csv")
    if not os.path.exists(md):
        pd.DataFrame({"Metric": ["mean", "stddev"], "Definition": ["arithmetic mean", "sample standard deviation"]}).to_csv(md, index=False)
    print("demo data written under", a.out)


if __name__ == "__main__":
    main()
