"""Doctests in public API modules (SURVEY.md §4 'pyspark doctests' row)."""

import doctest

import flowml.evaluation
import flowml.feature
import flowml.linalg
import flowml.pipeline
import flowml.tuning


def test_module_doctests():
    for mod in (flowml.pipeline, flowml.tuning, flowml.evaluation, flowml.linalg, flowml.feature):
        results = doctest.testmod(mod, verbose=False)
        assert results.failed == 0, f"{mod.__name__}: {results.failed} doctest failures"
