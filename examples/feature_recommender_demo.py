#!/usr/bin/env python3
"""Feature recommender (reference notebook feature_recommender_demo.ipynb):
explore the shipped corpus and map raw attributes to curated features."""

import pandas as pd

from _common import init_context  # noqa: F401  (keeps path setup)

from anovos_amd.feature_recommender import feature_explorer as fx
from anovos_amd.feature_recommender import feature_mapper as fm

print(fx.list_all_industry().head(5).to_string(index=False))
print(fx.list_feature_by_industry("telecommunication", num_of_feat=3).to_string(index=False))
attr = pd.DataFrame({"Attribute Name": ["cust_age", "avg_call_mins", "monthly_bill"],
                     "Attribute Description": ["age of the customer",
                                                "average call minutes per month",
                                                "monthly bill amount"]})
out = fm.feature_mapper(attr, name_column="Attribute Name",
                        desc_column="Attribute Description", top_n=2)
print(out.head(6).to_string(index=False))
