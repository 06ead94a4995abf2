# Copyright 2026 mlrun_amd authors
#
# Licensed under the Apache License, Version 2.0 (the "License");
# you may not use this file except in compliance with the License.
"""Client-side alerts package (reference mlrun/alerts/alert.py) and
project-level alert methods (reference project.py:4205)."""

import pytest

import mlrun_amd.common.schemas.alert as alert_objects
from mlrun_amd.alerts import AlertConfig


def _full_alert(project="p", name="drift"):
    return AlertConfig(
        project=project, name=name,
        summary="a drift was detected",
        severity=alert_objects.AlertSeverity.LOW,
        entities=alert_objects.EventEntities(
            kind=alert_objects.EventEntityKind.MODEL_ENDPOINT_RESULT,
            project=project, ids=["ep1.result"]),
        trigger=alert_objects.AlertTrigger(
            events=[alert_objects.EventKind.DATA_DRIFT_DETECTED]),
        criteria=alert_objects.AlertCriteria(count=3, period="1h"),
        notifications=[alert_objects.AlertNotification(
            notification={"kind": "console"})])


class TestAlertConfig:
    def test_to_dict_round_trip(self):
        alert = _full_alert()
        d = alert.to_dict()
        assert d["severity"] == "low"
        assert d["trigger"]["events"] == ["data-drift-detected"]
        assert d["entities"]["ids"] == ["ep1.result"]
        back = AlertConfig.from_dict(d)
        assert isinstance(back.trigger, alert_objects.AlertTrigger)
        assert back.criteria.count == 3

    def test_to_dict_requires_entities_and_notifications(self):
        from mlrun_amd.errors import MLRunBadRequestError

        with pytest.raises(MLRunBadRequestError):
            AlertConfig(project="p", name="n",
                        notifications=[alert_objects.AlertNotification(
                            notification={"kind": "console"})]).to_dict()
        with pytest.raises(MLRunBadRequestError):
            AlertConfig(
                project="p", name="n",
                entities=alert_objects.EventEntities(
                    kind=alert_objects.EventEntityKind.JOB,
                    project="p", ids=["r"])).to_dict()

    def test_template_seeding_user_wins(self):
        tmpl = {"summary": "tmpl summary", "severity": "high",
                "criteria": {"count": 5}, "reset_policy": "manual"}
        alert = AlertConfig(project="p", name="n", template=tmpl,
                            summary="mine")
        assert alert.summary == "mine"
        assert alert.criteria.count == 5
        assert alert.reset_policy == alert_objects.ResetPolicy.MANUAL

    def test_with_helpers_validate_types(self):
        alert = AlertConfig(project="p", name="n")
        with pytest.raises(ValueError):
            alert.with_entities({"kind": "job"})
        with pytest.raises(ValueError):
            alert.with_notifications([{"kind": "console"}])
        alert.with_entities(alert_objects.EventEntities(
            kind=alert_objects.EventEntityKind.JOB, project="p",
            ids=["r"]))
        alert.with_notifications([alert_objects.AlertNotification(
            notification={"kind": "console"})])
        assert alert.to_dict()["entities"]["kind"] == "job"

    def test_event_entity_validity_map(self):
        event = alert_objects.Event(
            kind=alert_objects.EventKind.FAILED,
            entity=alert_objects.EventEntities(
                kind=alert_objects.EventEntityKind.JOB,
                project="p", ids=["run1"]))
        assert event.is_valid()
        bad = alert_objects.Event(
            kind=alert_objects.EventKind.FAILED,
            entity=alert_objects.EventEntities(
                kind=alert_objects.EventEntityKind.MODEL_ENDPOINT_RESULT,
                project="p", ids=["x"]))
        assert not bad.is_valid()

    def test_alias_import(self):
        from mlrun.alerts import AlertConfig as Aliased

        assert Aliased is AlertConfig


class TestProjectAlertMethods:
    @pytest.fixture()
    def project(self, tmp_path, monkeypatch):
        import mlrun_amd
        from mlrun_amd.config import config

        monkeypatch.setattr(config, "base_dir", str(tmp_path))
        monkeypatch.setattr(config.httpdb, "dsn",
                            str(tmp_path / "alerts.db"), raising=False)
        mlrun_amd.db._run_db = None
        proj = mlrun_amd.new_project("alerts-proj",
                                     context=str(tmp_path))
        yield proj
        mlrun_amd.db._run_db = None

    def test_store_get_list_reset_delete(self, project):
        alert = _full_alert(project="alerts-proj")
        project.store_alert_config(alert)
        got = project.get_alert_config("drift")
        assert isinstance(got, AlertConfig)
        assert got.criteria.count == 3
        assert any(a["name"] == "drift"
                   for a in project.list_alerts_configs())
        project.reset_alert_config(alert_name="drift")
        project.delete_alert_config(alert_name="drift")
        assert all(a["name"] != "drift"
                   for a in project.list_alerts_configs())

    def test_name_mismatch_raises(self, project):
        alert = _full_alert(project="alerts-proj")
        with pytest.raises(ValueError):
            project.delete_alert_config(alert, "other-name")
        with pytest.raises(ValueError):
            project.reset_alert_config()
