"""Guard the non-imported scripts (tools/, bench.py, __graft_entry__)
against syntax rot — they only run on GPU boxes or as one-offs, so a
broken edit would otherwise surface at round end."""
import ast
import glob
import os

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_scripts_parse():
    files = (glob.glob(os.path.join(HERE, "tools", "**", "*.py"),
                       recursive=True)
             + [os.path.join(HERE, "bench.py"),
                os.path.join(HERE, "__graft_entry__.py"),
                os.path.join(HERE, "wukong_amd", "build.py")])
    assert len(files) >= 10
    for f in files:
        ast.parse(open(f).read(), filename=f)
