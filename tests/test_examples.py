"""Every example config in docs/examples renders and validates."""

import glob
import os

from containerpilot_amd import REPO_ROOT, native

EXAMPLES = sorted(glob.glob(os.path.join(REPO_ROOT, "docs", "examples",
                                         "*.json5")))


def test_examples_exist():
    assert len(EXAMPLES) >= 5


def test_examples_validate():
    for path in EXAMPLES:
        with open(path) as f:
            text = f.read()
        rendered = native.render_template(text)
        err = native.validate_config(rendered)
        assert err is None, "%s: %s" % (os.path.basename(path), err)
