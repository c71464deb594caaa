"""docs/API.md is generated — keep it in sync with the live class surface."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_api_doc_in_sync():
    sys.path.insert(0, os.path.join(REPO, "scripts"))
    import gen_api_doc

    fresh = gen_api_doc.main()
    committed = open(os.path.join(REPO, "docs", "API.md")).read()
    assert fresh == committed, (
        "docs/API.md is stale — run: python scripts/gen_api_doc.py"
    )
    assert fresh.count("### ") > 60  # the surface stays broad
