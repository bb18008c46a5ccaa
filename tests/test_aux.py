"""Aux runtime tests: telemetry, parameter server, launcher component,
futures timeouts."""

import json
import logging
import time
from datetime import timedelta

import pytest
import torch

from torchft_amd.futures import context_timeout, future_timeout
from torchft_amd.telemetry import JSONLineFormatter, setup_telemetry


class TestFutures:
    def test_future_timeout_fires(self):
        fut: torch.futures.Future = torch.futures.Future()
        timed = future_timeout(fut, timedelta(milliseconds=100))
        with pytest.raises(TimeoutError):
            timed.wait()

    def test_future_timeout_passthrough(self):
        fut: torch.futures.Future = torch.futures.Future()
        timed = future_timeout(fut, timedelta(seconds=5))
        fut.set_result(42)
        assert timed.wait() == 42

    def test_future_timeout_propagates_exception(self):
        fut: torch.futures.Future = torch.futures.Future()
        timed = future_timeout(fut, timedelta(seconds=5))
        fut.set_exception(ValueError("boom"))
        with pytest.raises(ValueError, match="boom"):
            timed.wait()

    def test_context_timeout_fires(self):
        fired = []
        with context_timeout(lambda: fired.append(1), timedelta(milliseconds=50)):
            time.sleep(0.3)
        assert fired == [1]

    def test_context_timeout_cancelled(self):
        fired = []
        with context_timeout(lambda: fired.append(1), timedelta(seconds=5)):
            pass
        time.sleep(0.1)
        assert fired == []


class TestTimerManagerLoad:
    def test_callback_exception_does_not_kill_timer_thread(self):
        import time as _time
        from datetime import timedelta

        from torchft_amd.futures import context_timeout

        fired = []

        def bad():
            raise RuntimeError("callback bug")

        # a raising callback must not wedge the shared timer loop
        try:
            with context_timeout(bad, timedelta(milliseconds=10)):
                _time.sleep(0.1)
        except RuntimeError:
            pass
        with context_timeout(lambda: fired.append(1), timedelta(milliseconds=20)):
            _time.sleep(0.2)
        assert fired == [1]

    def test_many_concurrent_future_timeouts(self):
        from datetime import timedelta

        import torch

        from torchft_amd.futures import future_timeout

        completed = []
        timed_out = []
        futs = []
        for i in range(40):
            f: torch.futures.Future = torch.futures.Future()
            tf = future_timeout(f, timedelta(milliseconds=150))
            futs.append((i, f, tf))
        # complete the even ones immediately; let the odd ones time out
        for i, f, _ in futs:
            if i % 2 == 0:
                f.set_result(i)
        for i, _, tf in futs:
            try:
                completed.append((i, tf.wait()))
            except TimeoutError:
                timed_out.append(i)
        assert [i for i, v in completed] == [i for i in range(40) if i % 2 == 0]
        assert all(v == i for i, v in completed)
        assert timed_out == [i for i in range(40) if i % 2 == 1]


class TestTelemetry:
    def test_json_formatter_includes_extras(self):
        fmt = JSONLineFormatter()
        record = logging.LogRecord(
            "torchft_quorums", logging.INFO, __file__, 1, "", (), None
        )
        record.quorum_id = 7
        record.replica_id = "rep0"
        out = json.loads(fmt.format(record))
        assert out["quorum_id"] == 7
        assert out["replica_id"] == "rep0"
        assert out["logger"] == "torchft_quorums"


class TestParameterServer:
    def test_session_roundtrip(self):
        from torchft_amd.parameter_server import ParameterServer
        from torchft_amd.process_group import ProcessGroup

        class EchoPS(ParameterServer):
            def forward(self, session_id: str, pg: ProcessGroup) -> None:
                t = torch.arange(8.0)
                pg.send([t], 1, tag=0).wait()

        ps = EchoPS()
        try:
            pg = EchoPS.connect(ps.address(), timeout=timedelta(seconds=10))
            t = torch.zeros(8)
            pg.recv([t], 0, tag=0).wait()
            torch.testing.assert_close(t, torch.arange(8.0))
        finally:
            ps.shutdown()

    def test_unknown_path_404(self):
        import urllib.error
        import urllib.request

        from torchft_amd.parameter_server import ParameterServer
        from torchft_amd.process_group import ProcessGroup

        class NopPS(ParameterServer):
            def forward(self, session_id: str, pg: ProcessGroup) -> None:
                pass

        ps = NopPS()
        try:
            with pytest.raises(urllib.error.HTTPError):
                urllib.request.urlopen(f"{ps.address()}/bogus", data=b"", timeout=10)
        finally:
            ps.shutdown()

    def test_two_concurrent_sessions(self):
        from concurrent.futures import ThreadPoolExecutor

        from torchft_amd.parameter_server import ParameterServer
        from torchft_amd.process_group import ProcessGroup

        class EchoPS(ParameterServer):
            def forward(self, session_id: str, pg: ProcessGroup) -> None:
                # echo the client's tensor back doubled
                t = torch.zeros(4)
                pg.recv([t], 1, tag=0).wait()
                pg.send([t * 2], 1, tag=1).wait()

        ps = EchoPS()
        try:
            def client(val: float) -> torch.Tensor:
                pg = EchoPS.connect(ps.address(), timeout=timedelta(seconds=20))
                pg.send([torch.full((4,), val)], 0, tag=0).wait()
                out = torch.zeros(4)
                pg.recv([out], 0, tag=1).wait()
                return out

            with ThreadPoolExecutor(2) as ex:
                f1 = ex.submit(client, 3.0)
                f2 = ex.submit(client, 5.0)
                torch.testing.assert_close(f1.result(timeout=60), torch.full((4,), 6.0))
                torch.testing.assert_close(f2.result(timeout=60), torch.full((4,), 10.0))
        finally:
            ps.shutdown()


class TestLauncherComponent:
    def test_build_replica_cmd(self):
        from torchft_amd.launcher import build_replica_cmd

        cmd, env = build_replica_cmd(
            ["train.py", "--steps", "10"], 1, 4, 2, "http://lh:123", 29650
        )
        assert "--nproc-per-node=2" in cmd
        assert cmd[-3:] == ["train.py", "--steps", "10"]
        assert env["REPLICA_GROUP_ID"] == "1"
        assert env["NUM_REPLICA_GROUPS"] == "4"
        assert env["TORCHFT_LIGHTHOUSE"] == "http://lh:123"


class TestTelemetrySetup:
    def test_file_handler_writes_json_lines(self, tmp_path):
        import json
        import logging

        from torchft_amd.telemetry import STRUCTURED_LOGGERS, setup_telemetry

        path = str(tmp_path / "telemetry.jsonl")
        saved = {n: list(logging.getLogger(n).handlers) for n in STRUCTURED_LOGGERS}
        try:
            setup_telemetry(path)
            log = logging.getLogger("torchft_quorums")
            log.info("", extra={"replica_id": "r0", "quorum_id": 3, "step": 5})
            for h in log.handlers:
                h.flush()
            lines = [json.loads(l) for l in open(path) if l.strip()]
            assert len(lines) == 1
            rec = lines[0]
            assert rec["logger"] == "torchft_quorums"
            assert rec["replica_id"] == "r0"
            assert rec["quorum_id"] == 3 and rec["step"] == 5
            assert "ts" in rec
        finally:
            for n in STRUCTURED_LOGGERS:
                lg = logging.getLogger(n)
                for h in list(lg.handlers):
                    if h not in saved[n]:
                        h.close()
                        lg.removeHandler(h)
                lg.propagate = True

    def test_unserializable_extra_becomes_repr(self):
        import json
        import logging

        from torchft_amd.telemetry import JSONLineFormatter

        rec = logging.LogRecord(
            "torchft_errors", logging.INFO, __file__, 1, "boom", (), None
        )
        rec.payload = object()  # not JSON-serializable
        out = json.loads(JSONLineFormatter().format(rec))
        assert out["msg"] == "boom"
        assert out["payload"].startswith("<object object")
