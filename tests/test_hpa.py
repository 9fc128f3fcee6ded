"""Unit tests for the HPA reconcile algorithm (reference semantics:
cuda-test-hpa.yaml:11-21, desired = ceil(current/target x replicas))."""

from mi355x_gpu_hpa.control import (
    HpaSpec,
    HpaState,
    MetricTarget,
    desired_replicas,
    desired_replicas_multi,
    reconcile,
    reconcile_multi,
)


def spec(**kw):
    defaults = dict(min_replicas=1, max_replicas=3, target_value=5.0)
    defaults.update(kw)
    return HpaSpec(**defaults)


class TestDesired:
    def test_within_tolerance_no_change(self):
        assert desired_replicas(spec(), 2, 5.2) == 2  # ratio 1.04 < 1.1
        assert desired_replicas(spec(), 2, 4.6) == 2  # ratio 0.92 > 0.9

    def test_scale_up_ceil(self):
        # the reference trigger: load ~8% vs target 5 on 1 replica
        assert desired_replicas(spec(), 1, 8.0) == 2
        assert desired_replicas(spec(), 1, 10.1) == 3

    def test_clamp_max(self):
        assert desired_replicas(spec(), 1, 100.0) == 3
        assert desired_replicas(spec(max_replicas=8), 1, 100.0) == 8

    def test_clamp_min(self):
        assert desired_replicas(spec(), 3, 0.0) == 1

    def test_missing_metric_no_change(self):
        assert desired_replicas(spec(), 2, None) == 2

    def test_scale_up_proportional(self):
        # 2 replicas at 7.5 avg: ratio 1.5 -> ceil(3.0) = 3
        assert desired_replicas(spec(), 2, 7.5) == 3


class TestReconcile:
    def test_scale_up_immediate(self):
        st = HpaState(current_replicas=1)
        assert reconcile(spec(), st, 20.0, now_s=0.0) == 3
        assert st.current_replicas == 3

    def test_scale_down_stabilized(self):
        s = spec(downscale_stabilization_s=300)
        st = HpaState(current_replicas=3)
        # load drops: desired=1, but within the window the max recommendation
        # (3, from the moment of the drop onwards only 1s) holds
        reconcile(s, st, 20.0, now_s=0.0)       # records desired=3
        assert reconcile(s, st, 0.0, now_s=10.0) == 3
        assert reconcile(s, st, 0.0, now_s=200.0) == 3
        # after the window expires, downscale lands
        assert reconcile(s, st, 0.0, now_s=301.0) == 1

    def test_flapping_suppressed(self):
        s = spec(downscale_stabilization_s=60)
        st = HpaState(current_replicas=2)
        reconcile(s, st, 10.0, now_s=0.0)   # up to 3... wait ratio 2 on 2 -> 3 (clamped)
        assert st.current_replicas == 3
        assert reconcile(s, st, 4.9, now_s=10.0) == 3   # dip within window
        assert reconcile(s, st, 10.0, now_s=20.0) == 3  # back up

    def test_multi_metric_max_wins(self):
        # deploy/multi-metric/cuda-test-hpa-multi.yaml semantics: desired =
        # max over metrics; a bandwidth-bound load scales on HBM BW before
        # busy% trips
        s = spec(max_replicas=8)
        metrics = [MetricTarget("cuda_test_gpu_avg", 40.0),
                   MetricTarget("cuda_test_hbm_bw_avg", 60.0)]
        # busy below target, bandwidth way above
        d = desired_replicas_multi(s, metrics, 1,
                                   {"cuda_test_gpu_avg": 30.0,
                                    "cuda_test_hbm_bw_avg": 95.0})
        assert d == 2  # ceil(95/60) = 2 > ceil(30/40)=1(tolerance band aside)

    def test_multi_metric_missing_one(self):
        s = spec(max_replicas=8)
        metrics = [MetricTarget("a", 10.0), MetricTarget("b", 10.0)]
        assert desired_replicas_multi(s, metrics, 2, {"a": None, "b": None}) == 2
        assert desired_replicas_multi(s, metrics, 2, {"a": 30.0}) == 6

    def test_reconcile_multi_stabilized(self):
        s = spec(max_replicas=8, downscale_stabilization_s=60)
        st = HpaState(current_replicas=1)
        metrics = [MetricTarget("a", 10.0)]
        assert reconcile_multi(s, st, metrics, {"a": 50.0}, now_s=0.0) == 5
        assert reconcile_multi(s, st, metrics, {"a": 0.0}, now_s=10.0) == 5
        assert reconcile_multi(s, st, metrics, {"a": 0.0}, now_s=61.0) == 1

    def test_v2_scaleup_policy_prevents_overshoot(self):
        """deploy/cuda-test-hpa.yaml's behavior block: 1 pod / 15 s. The
        reference (v2beta1) jumps straight to maxReplicas (README.md:123);
        the policy turns that into a stepped ramp."""
        s = spec(max_replicas=8, scale_up_pods=1, scale_up_period_s=15.0)
        st = HpaState(current_replicas=1)
        counts = []
        for i in range(8):
            counts.append(reconcile(s, st, 40.0, now_s=i * 15.0))
        assert counts == [2, 3, 4, 5, 6, 7, 8, 8]  # one pod per sync period

    def test_v2_scaleup_policy_within_period(self):
        s = spec(max_replicas=8, scale_up_pods=2, scale_up_period_s=15.0)
        st = HpaState(current_replicas=1)
        # two syncs inside one period add at most 2 pods total
        assert reconcile(s, st, 40.0, now_s=0.0) == 3
        assert reconcile(s, st, 40.0, now_s=5.0) == 3
        assert reconcile(s, st, 40.0, now_s=15.0) == 5

    def test_stabilized_replay_cannot_bypass_scaleup_policy(self):
        """behavior rate limits must apply AFTER
        stabilization (kube-controller-manager ordering). An earlier
        desired=8 that the scaleUp policy limited to 2 must not be
        replayed uncapped by the downscale-stabilization max on a later
        dip."""
        s = spec(max_replicas=8, scale_up_pods=1, scale_up_period_s=15.0,
                 downscale_stabilization_s=300.0)
        st = HpaState(current_replicas=1)
        # burst: desired=8, rate-limited to +1/period
        assert reconcile(s, st, 40.0, now_s=0.0) == 2
        # dip inside the stabilization window: the windowed max (8) is
        # replayed by stabilization but must still be rate-limited
        assert reconcile(s, st, 1.0, now_s=15.0) == 3
        assert reconcile(s, st, 1.0, now_s=30.0) == 4

    def test_scale_up_curve_to_eight(self):
        # config 4: 1 -> 8 replica scale-up under sustained high load
        s = spec(max_replicas=8)
        st = HpaState(current_replicas=1)
        seen = [st.current_replicas]
        for t in range(6):
            # each replica still saturated at 40% (target 5) => keep scaling
            reconcile(s, st, 40.0, now_s=float(t * 15))
            seen.append(st.current_replicas)
        assert st.current_replicas == 8
        assert seen[0] == 1 and sorted(seen) == seen  # monotone ramp


class TestScaleDownPolicy:
    def test_scaledown_pods_ramp(self):
        """v2 behavior.scaleDown Pods policy: at most N removed per
        period, applied after stabilization (upstream ordering)."""
        s = spec(max_replicas=8, downscale_stabilization_s=0.0,
                 scale_down_pods=1, scale_down_period_s=15.0)
        st = HpaState(current_replicas=8)
        counts = [reconcile(s, st, 1.0, now_s=i * 15.0) for i in range(8)]
        assert counts == [7, 6, 5, 4, 3, 2, 1, 1]

    def test_scaledown_within_period(self):
        s = spec(max_replicas=8, downscale_stabilization_s=0.0,
                 scale_down_pods=2, scale_down_period_s=15.0)
        st = HpaState(current_replicas=8)
        assert reconcile(s, st, 1.0, now_s=0.0) == 6
        assert reconcile(s, st, 1.0, now_s=5.0) == 6   # same period
        assert reconcile(s, st, 1.0, now_s=15.0) == 4

    def test_scaledown_policy_inactive_without_flag(self):
        s = spec(max_replicas=8, downscale_stabilization_s=0.0)
        st = HpaState(current_replicas=8)
        # unlimited drop straight to desired = ceil(8 * 1/5) = 2
        assert reconcile(s, st, 1.0, now_s=0.0) == 2
