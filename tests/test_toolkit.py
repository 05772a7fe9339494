"""Toolkit schema layer tests (hyperparameters, channels, metrics, errors).

Mirrors the reference test strategy for
test/unit/algorithm_toolkit/test_hyperparameter_validation.py.
"""
import pytest

from sagemaker_xgboost_container_amd.toolkit import channel_validation as cv
from sagemaker_xgboost_container_amd.toolkit import exceptions as exc
from sagemaker_xgboost_container_amd.toolkit import hyperparameter_validation as hpv
from sagemaker_xgboost_container_amd.toolkit import metrics as tkm


def _schema():
    return hpv.Hyperparameters(
        hpv.IntegerHyperparameter(name="int_hp", range=hpv.Interval(min_closed=1, max_closed=10), required=True),
        hpv.ContinuousHyperparameter(name="float_hp", range=hpv.Interval(min_open=0, max_open=1), default=0.5),
        hpv.CategoricalHyperparameter(name="cat_hp", range=["a", "b"], required=False),
        hpv.CommaSeparatedListHyperparameter(name="list_hp", range=["x", "y", "z"], required=False),
    )


class TestHyperparameters:
    def test_parse_and_defaults(self):
        out = _schema().validate({"int_hp": "3"})
        assert out == {"int_hp": 3, "float_hp": 0.5}

    def test_missing_required(self):
        with pytest.raises(exc.UserError, match="Missing required hyperparameter"):
            _schema().validate({})

    def test_extraneous(self):
        with pytest.raises(exc.UserError, match="Extraneous hyperparameter"):
            _schema().validate({"int_hp": "3", "bogus": "1"})

    def test_bad_parse(self):
        with pytest.raises(exc.UserError, match="could not parse"):
            _schema().validate({"int_hp": "three"})

    def test_range_violation(self):
        with pytest.raises(exc.UserError, match="is not in"):
            _schema().validate({"int_hp": "11"})
        with pytest.raises(exc.UserError):
            _schema().validate({"int_hp": "3", "float_hp": "1.0"})  # open max

    def test_categorical_and_list(self):
        out = _schema().validate({"int_hp": "1", "cat_hp": "a", "list_hp": "x,z"})
        assert out["cat_hp"] == "a"
        assert out["list_hp"] == ["x", "z"]
        with pytest.raises(exc.UserError):
            _schema().validate({"int_hp": "1", "list_hp": "x,q"})

    def test_alias(self):
        schema = _schema()
        schema.declare_alias("float_hp", "float_alias")
        out = schema.validate({"int_hp": "1", "float_alias": "0.25"})
        assert out["float_hp"] == 0.25

    def test_dependencies_order(self):
        @hpv.dependencies_validator(["base"])
        def need_base_big(value, deps):
            if deps.get("base", 0) < value:
                raise exc.UserError("dependent must be <= base")

        schema = hpv.Hyperparameters(
            hpv.IntegerHyperparameter(name="base", range=hpv.Interval(min_closed=0), default=5),
            hpv.IntegerHyperparameter(
                name="dep", range=hpv.Interval(min_closed=0), dependencies=need_base_big, required=False
            ),
        )
        assert schema.validate({"dep": "4"}) == {"base": 5, "dep": 4}
        with pytest.raises(exc.UserError, match="dependent must be"):
            schema.validate({"dep": "9"})

    def test_tuple_and_nested_list(self):
        schema = hpv.Hyperparameters(
            hpv.TupleHyperparameter(name="mono", range=[-1, 0, 1], required=False, default=None),
            hpv.NestedListHyperparameter(
                name="inter", range=hpv.Interval(min_closed=0), required=False, default=None
            ),
        )
        out = schema.validate({"mono": "(1, -1)", "inter": "[[0, 1], [2]]"})
        assert out["mono"] == (1, -1)
        assert out["inter"] == [[0, 1], [2]]
        with pytest.raises(exc.UserError):
            schema.validate({"mono": "(2,)"})

    def test_format(self):
        spec = _schema().format()
        int_spec = next(s for s in spec if s["Name"] == "int_hp")
        assert int_spec["Type"] == "Integer"
        assert int_spec["IsRequired"] is True
        assert int_spec["Range"]["IntegerParameterRangeSpecification"] == {"MinValue": "1", "MaxValue": "10"}


class TestInterval:
    def test_membership(self):
        iv = hpv.Interval(min_open=0, max_closed=1)
        assert 0 not in iv and 0.5 in iv and 1 in iv and 1.5 not in iv
        assert str(iv) == "(0, 1]"

    def test_conflicting_bounds(self):
        with pytest.raises(exc.AlgorithmError):
            hpv.Interval(min_open=0, min_closed=0)


class TestChannels:
    def _channels(self):
        train = cv.Channel(name="train", required=True)
        train.add("csv", cv.Channel.FILE_MODE, cv.Channel.REPLICATED)
        val = cv.Channel(name="validation", required=False)
        val.add("csv", cv.Channel.FILE_MODE, cv.Channel.REPLICATED)
        chans = cv.Channels(train, val)
        chans.set_default_content_type("csv")
        return chans

    def test_ok(self):
        cfg = {"train": {"TrainingInputMode": "File", "S3DistributionType": "FullyReplicated"}}
        out = self._channels().validate(cfg)
        assert out["train"]["ContentType"] == "csv"

    def test_missing_required(self):
        with pytest.raises(exc.UserError, match="Missing required channel"):
            self._channels().validate({})

    def test_unsupported_combo(self):
        cfg = {
            "train": {
                "ContentType": "csv",
                "TrainingInputMode": "Pipe",
                "S3DistributionType": "FullyReplicated",
            }
        }
        with pytest.raises(exc.UserError, match="not supported"):
            self._channels().validate(cfg)

    def test_extraneous_channel(self):
        cfg = {
            "train": {"ContentType": "csv", "TrainingInputMode": "File", "S3DistributionType": "FullyReplicated"},
            "bogus": {"ContentType": "csv", "TrainingInputMode": "File", "S3DistributionType": "FullyReplicated"},
        }
        with pytest.raises(exc.UserError, match="Extraneous channel"):
            self._channels().validate(cfg)


class TestMetricsAndExceptions:
    def test_metric_requires_direction(self):
        with pytest.raises(exc.AlgorithmError):
            tkm.Metric(name="m", regex="r", tunable=True)

    def test_metrics_formats(self):
        ms = tkm.Metrics(
            tkm.Metric(name="a", regex="ra", direction=tkm.Metric.MAXIMIZE),
            tkm.Metric(name="b", regex="rb", tunable=False),
        )
        assert ms.names == ["a", "b"]
        assert ms.format_tunable() == [{"MetricName": "a", "Type": "Maximize"}]
        assert {"Name": "b", "Regex": "rb"} in ms.format_definitions()

    def test_exception_formatting(self):
        e = exc.UserError("boom", caused_by=ValueError("inner"))
        assert "boom" in e.message and "ValueError" in e.message
        e2 = exc.AlgorithmError(caused_by=ValueError("inner"))
        assert "inner" in e2.message


class TestIntegrationLogging:
    """`[ts:LEVEL] msg` console format (reference integration.py:16-52) —
    the format CloudWatch scrapes eval lines from."""

    def test_format_string(self):
        import logging
        import re

        from sagemaker_xgboost_container_amd.algorithm_mode import integration

        fmt = integration.LOGGING_CONFIG["formatters"]["standard"]
        record = logging.LogRecord("x", logging.INFO, "f.py", 1, "hello eval", None, None)
        formatter = logging.Formatter(fmt["format"], datefmt=fmt.get("datefmt"))
        line = formatter.format(record)
        assert re.match(r"^\[\d{4}-\d{2}-\d{2}:\d{2}:\d{2}:\d{2}:INFO\] hello eval$", line), line

    def test_setup_returns_named_logger(self):
        from sagemaker_xgboost_container_amd.algorithm_mode import integration

        lg = integration.setup_main_logger("smoke-test-logger")
        assert lg.name == "smoke-test-logger"
