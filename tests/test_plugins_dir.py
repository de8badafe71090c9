"""Out-of-tree plugin loading (reference --plugins-dir dlopen,
framework/plugins.go): a module dropped in a directory self-registers
and is usable from conf tiers by name."""


def test_load_plugins_dir(tmp_path):
    plug = tmp_path / "fifo_boost.py"
    plug.write_text(
        "from volcano_amd.scheduler.plugins.base import Plugin, register\n"
        "\n"
        "@register('fifo-boost')\n"
        "class FifoBoost(Plugin):\n"
        "    def on_session_open(self, ssn):\n"
        "        ssn.add_job_order_fn(\n"
        "            lambda a, b: (a.creation_timestamp >\n"
        "                          b.creation_timestamp)\n"
        "                         - (a.creation_timestamp <\n"
        "                            b.creation_timestamp),\n"
        "            key=lambda j: j.creation_timestamp)\n")
    from volcano_amd.scheduler.plugins import (PLUGIN_REGISTRY,
                                               load_plugins_dir, new_plugin)
    loaded = load_plugins_dir(str(tmp_path))
    assert loaded == ["fifo-boost"]
    assert "fifo-boost" in PLUGIN_REGISTRY
    p = new_plugin("fifo-boost")
    assert p.name == "fifo-boost"

    # usable end-to-end from a conf tier
    from volcano_amd.utils import synth
    from volcano_amd.utils.uthelper import TestCommonStruct
    t = TestCommonStruct(
        podgroups=[],
        queues=[synth.make_queue("default")],
        nodes=[synth.make_node("n1")],
        tiers=[["fifo-boost", "gang"]],
        actions=["enqueue", "allocate"],
    ).run()
    assert t.binder.binds == {}
