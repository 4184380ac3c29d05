"""showflakes-equivalent pytest plugin.

The reference depends on an external `showflakes` plugin (empty git
submodule) whose interface is recoverable from its call sites
(reference experiment.py:153-158) and output parser (:260-277):

  --record-file=PATH  write one line per executed test: "outcome\\tnodeid"
                      (failure detection downstream is substring-based:
                      "failed" in outcome)
  --shuffle           randomize test execution order
  --set-exitstatus    a run whose tests merely FAIL must still count as a
                      successful container run (failures are the data the
                      study collects), so ordinary test failures are
                      coerced to exit status 0; collection errors and
                      internal errors keep their nonzero status.

Register with `-p flake16_framework_amd.collect.showflakes`.
"""

import random


def pytest_addoption(parser):
    group = parser.getgroup("showflakes")
    group.addoption("--record-file", action="store", default=None,
                    help="TSV file recording outcome per test")
    group.addoption("--shuffle", action="store_true", default=False,
                    help="randomize test order")
    group.addoption("--set-exitstatus", action="store_true", default=False,
                    help="exit 0 when tests ran (failures are data)")


def pytest_collection_modifyitems(config, items):
    if config.getoption("--shuffle"):
        random.shuffle(items)


class _Recorder:
    """Registered as a per-session pytest plugin only when --record-file is
    active, so runs without the option pay nothing and two sessions in one
    process cannot leak reports into each other."""

    def __init__(self, path):
        self.path = path
        self.outcomes = {}

    def pytest_runtest_logreport(self, report):
        if report.when not in ("setup", "call", "teardown"):
            return
        nid = report.nodeid
        prev = self.outcomes.get(nid)
        if report.outcome == "failed" or prev == "failed":
            outcome = "failed"
        elif report.outcome == "skipped" or prev == "skipped":
            outcome = "skipped"
        else:
            outcome = "passed"
        self.outcomes[nid] = outcome

    def pytest_sessionfinish(self, session, exitstatus):
        with open(self.path, "w") as fd:
            for nid, outcome in self.outcomes.items():
                fd.write(f"{outcome}\t{nid}\n")


def pytest_configure(config):
    path = config.getoption("--record-file")
    if path:
        rec = _Recorder(path)
        config._showflakes_recorder = rec
        config.pluginmanager.register(rec)


def pytest_unconfigure(config):
    rec = getattr(config, "_showflakes_recorder", None)
    if rec is not None:
        config.pluginmanager.unregister(rec)
        del config._showflakes_recorder


def pytest_sessionfinish(session, exitstatus):
    if session.config.getoption("--set-exitstatus"):
        # ordinary test failures are data, not an orchestration error
        if exitstatus == 1:  # pytest.ExitCode.TESTS_FAILED
            session.exitstatus = 0
