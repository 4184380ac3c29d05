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
    def __init__(self, path):
        self.path = path
        self.outcomes = {}

    def record(self, report):
        nid = report.nodeid
        prev = self.outcomes.get(nid)
        if report.outcome == "failed" or prev == "failed":
            outcome = "failed"
        elif report.outcome == "skipped" or prev == "skipped":
            outcome = "skipped"
        else:
            outcome = "passed"
        self.outcomes[nid] = outcome

    def flush(self):
        with open(self.path, "w") as fd:
            for nid, outcome in self.outcomes.items():
                fd.write(f"{outcome}\t{nid}\n")


def pytest_configure(config):
    path = config.getoption("--record-file")
    if path:
        config._showflakes_recorder = _Recorder(path)


def pytest_runtest_logreport(report):
    import pytest  # noqa: F401
    # recorder reached via the config on the session; stored at configure
    # time — pytest passes report without config, so use the plugin trick:
    _report_sink.append(report)


_report_sink = []


def pytest_sessionfinish(session, exitstatus):
    config = session.config
    rec = getattr(config, "_showflakes_recorder", None)
    if rec is not None:
        for report in _report_sink:
            if report.when in ("setup", "call", "teardown"):
                rec.record(report)
        rec.flush()
    _report_sink.clear()

    if config.getoption("--set-exitstatus"):
        # ordinary test failures are data, not an orchestration error
        if exitstatus == 1:  # pytest.ExitCode.TESTS_FAILED
            session.exitstatus = 0
