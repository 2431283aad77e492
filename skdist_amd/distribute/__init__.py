"""
Distributed meta-estimators — the product layer (reference SURVEY.md §1 L2).
"""
