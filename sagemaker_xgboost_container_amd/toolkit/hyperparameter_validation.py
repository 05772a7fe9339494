"""Declarative typed hyperparameter schema with validation.

SageMaker delivers all hyperparameters as strings. This module provides the
typed schema layer: each hyperparameter declares a type (parse), an optional
range, optional cross-field dependency validation, required/default
semantics, alias names, and SageMaker CreateAlgorithm API formatting.

``Hyperparameters.validate`` runs four phases:
  0. alias substitution,
  1. required-check / default fill,
  2. typed parse,
  3. range validation,
  4. dependency validation in topological order of declared dependencies.

Behavior parity with the reference toolkit
(sagemaker_algorithm_toolkit/hyperparameter_validation.py:19-433); the
implementation here is original.
"""
import ast
import sys

from . import exceptions as exc


class Range:
    """Interface for a hyperparameter's legal-value set."""

    def __contains__(self, value):
        raise NotImplementedError

    def format(self):
        raise NotImplementedError

    def __str__(self):
        raise NotImplementedError


class Interval(Range):
    """A numeric interval with independently open/closed endpoints."""

    LINEAR_SCALE = "Linear"

    def __init__(self, min_open=None, min_closed=None, max_open=None, max_closed=None, scale=None):
        if min_open is not None and min_closed is not None:
            raise exc.AlgorithmError("Only one of min_open and min_closed can be set")
        if max_open is not None and max_closed is not None:
            raise exc.AlgorithmError("Only one of max_open and max_closed can be set")
        self.min_open = min_open
        self.min_closed = min_closed
        self.max_open = max_open
        self.max_closed = max_closed
        self.scale = scale

    def __contains__(self, value):
        if self.min_open is not None and not value > self.min_open:
            return False
        if self.min_closed is not None and not value >= self.min_closed:
            return False
        if self.max_open is not None and not value < self.max_open:
            return False
        if self.max_closed is not None and not value <= self.max_closed:
            return False
        return True

    def __str__(self):
        if self.min_open is not None:
            lo = f"({self.min_open}, "
        elif self.min_closed is not None:
            lo = f"[{self.min_closed}, "
        else:
            lo = "(-inf, "
        if self.max_open is not None:
            hi = f"{self.max_open})"
        elif self.max_closed is not None:
            hi = f"{self.max_closed}]"
        else:
            hi = "+inf)"
        return lo + hi

    def _bound(self, open_end, closed_end, fallback):
        if open_end is not None:
            return str(open_end)
        if closed_end is not None:
            return str(closed_end)
        return str(fallback)

    def format_as_integer(self):
        return (
            self._bound(self.min_open, self.min_closed, -(2**31)),
            self._bound(self.max_open, self.max_closed, 2**31 - 1),
        )

    def format_as_continuous(self):
        return (
            self._bound(self.min_open, self.min_closed, -sys.float_info.max),
            self._bound(self.max_open, self.max_closed, sys.float_info.max),
        )


class Hyperparameter:
    """One declared hyperparameter: name, type, range, deps, default."""

    def __init__(
        self,
        name,
        range=None,
        dependencies=None,
        required=None,
        default=None,
        tunable=False,
        tunable_recommended_range=None,
    ):
        if required is None and default is None:
            raise exc.AlgorithmError("At least one of 'required' or 'default' must be specified.")
        self.name = name
        self.range = range
        self.dependencies = dependencies
        self.required = required
        self.default = default
        self.tunable = tunable
        self.tunable_recommended_range = tunable_recommended_range

    @property
    def type(self):
        return "FreeText"

    def parse(self, value):
        return value

    def validate_range(self, value):
        if self.range is not None and value not in self.range:
            raise exc.UserError(f"Hyperparameter {self.name}: {value} is not in {self.range}")

    def validate_dependencies(self, value, dependencies):
        if self.dependencies is not None:
            self.dependencies(value, dependencies)

    def format_range(self):
        raise NotImplementedError

    def format_tunable_range(self):
        return None

    def format(self):
        spec = {
            "Name": self.name,
            "Description": self.name,
            "Type": self.type,
            "IsTunable": self.tunable,
            "IsRequired": bool(self.required),
        }
        try:
            spec["Range"] = self.format_range()
        except NotImplementedError:
            pass
        if self.default is not None:
            spec["DefaultValue"] = str(self.default)
        return spec


class _RangeRequired(Hyperparameter):
    """Base for hyperparameter types for which a range is mandatory."""

    def __init__(self, *args, **kwargs):
        if kwargs.get("range") is None:
            raise exc.AlgorithmError("range must be specified")
        super().__init__(*args, **kwargs)


class IntegerHyperparameter(_RangeRequired):
    @property
    def type(self):
        return "Integer"

    def parse(self, value):
        return int(value)

    def format_range(self):
        lo, hi = self.range.format_as_integer()
        return {"IntegerParameterRangeSpecification": {"MinValue": lo, "MaxValue": hi}}

    def format_tunable_range(self):
        if not self.tunable or self.tunable_recommended_range is None:
            return None
        lo, hi = self.tunable_recommended_range.format_as_integer()
        return {
            "IntegerParameterRanges": [
                {
                    "MinValue": lo,
                    "MaxValue": hi,
                    "Name": self.name,
                    "ScalingType": self.tunable_recommended_range.scale,
                }
            ]
        }


class ContinuousHyperparameter(_RangeRequired):
    @property
    def type(self):
        return "Continuous"

    def parse(self, value):
        return float(value)

    def format_range(self):
        lo, hi = self.range.format_as_continuous()
        return {"ContinuousParameterRangeSpecification": {"MinValue": lo, "MaxValue": hi}}

    def format_tunable_range(self):
        if not self.tunable or self.tunable_recommended_range is None:
            return None
        lo, hi = self.tunable_recommended_range.format_as_continuous()
        return {
            "ContinuousParameterRanges": [
                {
                    "Name": self.name,
                    "MinValue": lo,
                    "MaxValue": hi,
                    "ScalingType": self.tunable_recommended_range.scale,
                }
            ]
        }


class CategoricalHyperparameter(_RangeRequired):
    @property
    def type(self):
        return "Categorical"

    def _range_values(self, range_):
        if isinstance(range_, (list, tuple)):
            return list(range_)
        return range_.format()

    def format_range(self):
        return {"CategoricalParameterRangeSpecification": {"Values": self._range_values(self.range)}}

    def format_tunable_range(self):
        if not self.tunable or self.tunable_recommended_range is None:
            return None
        return {
            "CategoricalParameterRanges": [
                {"Name": self.name, "Values": self._range_values(self.tunable_recommended_range)}
            ]
        }


class CommaSeparatedListHyperparameter(_RangeRequired):
    def parse(self, value):
        return value.split(",")

    def validate_range(self, value):
        for item in value:
            if item not in self.range:
                raise exc.UserError(f"Hyperparameter {self.name}: value {value} not in range {self.range}")


class NestedListHyperparameter(_RangeRequired):
    """A list of lists, e.g. interaction_constraints '[[0,1],[2,3,4]]'."""

    def parse(self, value):
        if isinstance(value, str):
            return ast.literal_eval(value)
        return value

    def format_range(self):
        lo, hi = self.range.format_as_integer()
        return {"NestedParameterRangeSpecification": {"MinValue": lo, "MaxValue": hi}}

    def validate_range(self, value):
        for inner in value:
            for item in inner:
                if item not in self.range:
                    raise exc.UserError(f"Hyperparameter {self.name}: value {value} not in range {self.range}")


class TupleHyperparameter(_RangeRequired):
    """A tuple, e.g. monotone_constraints '(0,1,-1)'."""

    def parse(self, value):
        if isinstance(value, str):
            return ast.literal_eval(value)
        return value

    def format_range(self):
        return {"TupleParameterRangeSpecification": {"Values": self.range}}

    def validate_range(self, value):
        for item in value:
            if item not in self.range:
                raise exc.UserError(f"Hyperparameter {self.name}: value {value} not in range {self.range}")


class Hyperparameters:
    """A full schema: an ordered collection of Hyperparameter declarations."""

    def __init__(self, *hyperparameters):
        self.hyperparameters = {hp.name: hp for hp in hyperparameters}
        self.aliases = {}

    def declare_alias(self, key_name, alias_name):
        if key_name not in self.hyperparameters:
            raise exc.AlgorithmError(f"Key name {key_name}: does not exist in list of hyperparameters")
        self.aliases[alias_name] = key_name

    def __getitem__(self, name):
        return self.hyperparameters[name]

    def _canonicalize(self, user_values):
        return {self.aliases.get(name, name): value for name, value in user_values.items()}

    def _dependency_order(self, names):
        """Names ordered so every hyperparameter follows its dependencies."""
        order = []
        seen = set()

        def visit(name):
            seen.add(name)
            deps = self.hyperparameters[name].dependencies
            if deps:
                for dep in deps:
                    if dep in names and dep not in seen:
                        visit(dep)
            order.append(name)

        for name in names:
            if name not in seen:
                visit(name)
        return order

    def validate(self, user_hyperparameters):
        # Phase 0: map alias names onto canonical keys.
        values = self._canonicalize(dict(user_hyperparameters))

        # Phase 1: required-check / default fill.
        for name, hp in self.hyperparameters.items():
            if name not in values:
                if hp.required:
                    raise exc.UserError(f"Missing required hyperparameter: {name}")
                if hp.default is not None:
                    values[name] = hp.default

        # Phase 2: typed parse.
        parsed = {}
        for name, raw in values.items():
            if name not in self.hyperparameters:
                raise exc.UserError(f"Extraneous hyperparameter found: {name}")
            try:
                parsed[name] = self.hyperparameters[name].parse(raw)
            except (ValueError, SyntaxError) as e:
                raise exc.UserError(f"Hyperparameter {name}: could not parse value", caused_by=e)

        # Phase 3: range validation.
        for name, value in parsed.items():
            try:
                self.hyperparameters[name].validate_range(value)
            except exc.UserError:
                raise
            except Exception as e:
                raise exc.AlgorithmError(
                    f"Hyperparameter {name}: unexpected failure when validating {value}", caused_by=e
                )

        # Phase 4: dependency validation, dependencies first.
        validated = {}
        for name in self._dependency_order(list(parsed)):
            hp = self.hyperparameters[name]
            if hp.dependencies:
                dep_values = {d: validated[d] for d in hp.dependencies if d in validated}
                hp.validate_dependencies(parsed[name], dep_values)
            validated[name] = parsed[name]
        return validated

    def format(self):
        return [hp.format() for hp in self.hyperparameters.values()]


class range_validator:
    """Decorator: wrap ``f(range, value) -> bool`` as a Range object."""

    def __init__(self, range):
        self.range = range

    def __call__(self, f):
        outer = self

        class _FunctionRange(Range):
            def format(self):
                return outer.range

            def __str__(self):
                return str(outer.range)

            def __contains__(self, value):
                return f(outer.range, value)

        return _FunctionRange()


class dependencies_validator:
    """Decorator: wrap ``f(value, dependencies)`` as an iterable validator."""

    def __init__(self, dependencies):
        self.dependencies = dependencies

    def __call__(self, f):
        outer = self

        class _DependencyValidator:
            def __init__(self):
                self.dependencies = outer.dependencies

            def __iter__(self):
                return iter(self.dependencies)

            def __call__(self, value, dependencies):
                return f(value, dependencies)

        return _DependencyValidator()
