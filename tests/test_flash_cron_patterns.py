"""Unit tests: Flash autoscaler decisions, cron evaluation, file patterns
(SURVEY §4.7 unit-level strategy: decision logic tested directly)."""

from __future__ import annotations

from datetime import datetime

from modal_amd.flash import FlashAutoscaler, FlashManager
from modal_amd.scheduler.cron import cron_matches


class TestFlashAutoscaler:
    def test_proportional_scale_up_after_window(self):
        metric = {"v": 100.0}
        a = FlashAutoscaler(
            lambda: metric["v"], target_value=50.0,
            min_replicas=1, max_replicas=8,
            scale_up_stabilization=10.0, scale_down_stabilization=300.0,
        )
        # metric 2x target -> desired doubles, but only after 10 s of persistence
        assert a.decide(2, now=1000.0) == 2
        assert a.decide(2, now=1005.0) == 2
        assert a.decide(2, now=1011.0) == 4

    def test_scale_down_uses_longer_window(self):
        metric = {"v": 10.0}
        a = FlashAutoscaler(
            lambda: metric["v"], target_value=50.0,
            scale_up_stabilization=0.0, scale_down_stabilization=300.0,
        )
        assert a.decide(4, now=0.0) == 4       # pending down, not yet applied
        assert a.decide(4, now=200.0) == 4
        assert a.decide(4, now=301.0) == 1     # ceil(4 * 0.2) = 1

    def test_tolerance_band_holds_steady(self):
        a = FlashAutoscaler(lambda: 52.0, target_value=50.0, tolerance=0.1)
        assert a.compute_desired(3, 52.0) == 3

    def test_clamped_to_bounds(self):
        a = FlashAutoscaler(lambda: 1000.0, target_value=1.0, max_replicas=6)
        assert a.compute_desired(4, 1000.0) == 6
        a2 = FlashAutoscaler(lambda: 0.0, target_value=1.0, min_replicas=2)
        assert a2.compute_desired(4, 0.001) == 2

    def test_direction_flip_resets_window(self):
        metric = {"v": 100.0}
        a = FlashAutoscaler(
            lambda: metric["v"], target_value=50.0,
            scale_up_stabilization=10.0, scale_down_stabilization=10.0,
        )
        assert a.decide(2, now=0.0) == 2       # pending up since t=0
        metric["v"] = 10.0                     # direction flips: down
        assert a.decide(2, now=5.0) == 2       # new pending, window restarts
        assert a.decide(2, now=11.0) == 2      # only 6 s into the down window
        assert a.decide(2, now=16.0) == 1

    def test_registry(self):
        m = FlashManager()
        ep = m.register("svc", "http://127.0.0.1:9")
        assert [e.name for e in m.list()] == ["svc"]
        assert ep.url.endswith(":9")
        m.deregister("svc")
        assert m.list() == []


class TestCron:
    def test_exact_minute(self):
        assert cron_matches("30 14 * * *", datetime(2026, 9, 13, 14, 30))
        assert not cron_matches("30 14 * * *", datetime(2026, 9, 13, 14, 31))

    def test_step_and_range(self):
        assert cron_matches("*/15 * * * *", datetime(2026, 9, 13, 8, 45))
        assert not cron_matches("*/15 * * * *", datetime(2026, 9, 13, 8, 50))
        assert cron_matches("0 9-17 * * *", datetime(2026, 9, 13, 12, 0))
        assert not cron_matches("0 9-17 * * *", datetime(2026, 9, 13, 20, 0))

    def test_weekday_and_lists(self):
        # 2026-09-13 is a Sunday (cron dow 0)
        assert cron_matches("0 0 * * 0", datetime(2026, 9, 13, 0, 0))
        assert not cron_matches("0 0 * * 1-5", datetime(2026, 9, 13, 0, 0))
        assert cron_matches("0 0 * * 1,3,5", datetime(2026, 9, 14, 0, 0))  # Monday

    def test_posix_dom_dow_or_semantics(self):
        # both dom and dow restricted -> OR (POSIX cron): fires on the 1st,
        # the 15th, AND every Monday
        expr = "0 0 1,15 * 1"
        assert cron_matches(expr, datetime(2026, 9, 1, 0, 0))  # 1st (a Tuesday)
        assert cron_matches(expr, datetime(2026, 9, 15, 0, 0))  # 15th
        assert cron_matches(expr, datetime(2026, 9, 14, 0, 0))  # a Monday, not 1st/15th
        assert not cron_matches(expr, datetime(2026, 9, 13, 0, 0))  # Sunday the 13th
        # one field unrestricted -> AND as before
        assert cron_matches("0 0 13 * *", datetime(2026, 9, 13, 0, 0))
        assert not cron_matches("0 0 * * 1", datetime(2026, 9, 13, 0, 0))


class TestFilePatterns:
    def test_basic_globs(self):
        from modal_amd.file_pattern_matcher import FilePatternMatcher

        m = FilePatternMatcher("**/*.py", "!**/test_*.py")
        assert m("pkg/mod.py")
        assert not m("pkg/test_mod.py")
        assert not m("pkg/data.csv")


class TestFlashLoop:
    """End-to-end Flash wiring (round-1 review Missing #6): the loop
    scrapes a metric, decides, and APPLIES replica changes on the pool."""

    def test_scale_up_then_down_with_windows(self, client):
        import asyncio

        import modal_amd as modal
        from modal_amd._sync import synchronizer
        from modal_amd.flash import FlashAutoscalerLoop

        app = modal.App("flash-loop")

        @app.function()
        def work(x):
            return x

        with app.run(client=client):
            work.remote(1)  # force a worker + function row
            svc = client.svc
            fid = work.object_id
            load = {"value": 8.0}
            loop = FlashAutoscalerLoop(
                svc, fid, target_value=2.0,
                get_metric=lambda: load["value"],
                min_replicas=1, max_replicas=4,
                scale_up_stabilization=0.0,
                scale_down_stabilization=0.5,
                tolerance=0.1,
            )

            async def run_up():
                before = loop.current_replicas()
                applied = await loop.tick()
                # metric 4x target -> scale toward max within the up window
                deadline = asyncio.get_event_loop().time() + 30
                while loop.current_replicas() < min(4, before + 1):
                    if asyncio.get_event_loop().time() > deadline:
                        raise AssertionError("scale-up never applied")
                    await asyncio.sleep(0.1)
                return applied, loop.current_replicas()

            applied, now_replicas = synchronizer.run(run_up())
            assert applied >= 2 and now_replicas >= 2

            # drop the load: scale-down only after the down window elapses
            load["value"] = 0.1

            async def run_down():
                import time as _t

                n0 = loop.current_replicas()
                first = await loop.tick()  # opens the down window
                assert first == n0, "scale-down applied before its window"
                await asyncio.sleep(0.7)
                second = await loop.tick()
                assert second < n0, "scale-down never applied after window"

            synchronizer.run(run_down())

    def test_prometheus_scrape_parses_text(self):
        import http.server
        import threading

        from modal_amd.flash import scrape_prometheus_metric

        body = (
            b"# HELP queue_depth depth\n"
            b"# TYPE queue_depth gauge\n"
            b'queue_depth{partition="a"} 3\n'
            b'queue_depth{partition="b"} 4\n'
            b"queue_depth_total 99\n"
            b"other_metric 7\n"
        )

        class H(http.server.BaseHTTPRequestHandler):
            def do_GET(self):
                self.send_response(200)
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a):
                pass

        server = http.server.HTTPServer(("127.0.0.1", 0), H)
        threading.Thread(target=server.serve_forever, daemon=True).start()
        try:
            url = f"http://127.0.0.1:{server.server_port}/_metrics"
            assert scrape_prometheus_metric(url, "queue_depth") == 7.0
            assert scrape_prometheus_metric(url, "other_metric") == 7.0
            assert scrape_prometheus_metric(url, "missing") is None
        finally:
            server.shutdown()
