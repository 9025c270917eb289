"""APIFields: the CRD spec field tree built from field markers.

Parity target: reference internal/workload/v1/kinds/api.go (AddField
:33-90, GenerateAPISpec :92-116, GenerateSampleSpec :118-136, kubebuilder
default markers :264-277).  Dotted marker paths create intermediate
structs; conflicting redefinitions raise; the tree renders both the Go
Spec struct set and the sample manifest spec.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

from ..utils import go_title
from .markers import FieldType

from ..errors import OperatorBuilderError


class APIFieldError(OperatorBuilderError):
    pass


ERR_OVERWRITE = "an attempt to overwrite existing value was made"


def _go_fmt_value(value: Any) -> str:
    """Go fmt %v rendering for the non-scalar values a manifest can
    carry (reference api.go getSampleValue uses fmt.Sprintf("%v", ...)):
    slices render as [a b c], maps as map[k:v]."""
    if isinstance(value, bool):
        return "true" if value else "false"
    if isinstance(value, (list, tuple)):
        return "[" + " ".join(_go_fmt_value(v) for v in value) + "]"
    if isinstance(value, dict):
        inner = " ".join(
            f"{k}:{_go_fmt_value(v)}" for k, v in sorted(value.items())
        )
        return f"map[{inner}]"
    return str(value)


@dataclass
class APIFields:
    name: str
    type: FieldType
    tags: str = ""
    struct_name: str = ""
    manifest_name: str = ""
    comments: list[str] = field(default_factory=list)
    markers: list[str] = field(default_factory=list)
    children: list["APIFields"] = field(default_factory=list)
    default: str = ""
    sample: str = ""
    last: bool = False

    # ---- building ------------------------------------------------------

    def add_field(
        self,
        path: str,
        field_type: FieldType,
        comments: Optional[list[str]],
        sample: Any,
        has_default: bool,
    ) -> None:
        obj = self
        parts = path.split(".")
        last = parts[-1]

        for part in parts[:-1]:
            found = None
            for child in obj.children:
                if child.manifest_name == part:
                    if child.type != FieldType.STRUCT:
                        raise APIFieldError(
                            f"{ERR_OVERWRITE} for api field {path}"
                        )
                    found = child
                    break

            if found is None:
                child = obj.new_child(part, FieldType.STRUCT, sample)
                child.markers.append("+kubebuilder:validation:Optional")
                child.generate_struct_name(path)
                obj.children.append(child)
                obj = child
            else:
                obj = found

        new_child = obj.new_child(last, field_type, sample)
        new_child.last = True
        new_child.set_comments_and_default(comments, sample, has_default)

        for child in obj.children:
            if child.manifest_name == last:
                if not child.is_equal(new_child):
                    raise APIFieldError(
                        f"{ERR_OVERWRITE} for api field {path}"
                    )
                child.set_comments_and_default(comments, sample, has_default)
                return

        obj.children.append(new_child)

    def new_child(
        self, name: str, field_type: FieldType, sample: Any
    ) -> "APIFields":
        child = APIFields(
            name=go_title(name),
            manifest_name=name,
            type=field_type,
            tags=f'`json:"{name},omitempty"`',
            comments=[],
            markers=[],
        )
        child.set_sample(sample)
        return child

    def generate_struct_name(self, path: str) -> None:
        parts = ["Spec"]
        for part in path.split("."):
            parts.append(go_title(part))
            if part == self.manifest_name:
                break
        self.struct_name = "".join(parts)

    def is_equal(self, other: "APIFields") -> bool:
        if self.type != other.type:
            return False
        if (
            self.default == ""
            or self.default == other.default
            or other.default == ""
        ):
            if len(self.comments) == 0 or len(other.comments) == 0:
                return True
            if len(self.comments) == len(other.comments):
                return self.comments == other.comments
        return False

    # ---- samples and defaults -----------------------------------------

    def get_sample_value(self, sample_val: Any) -> str:
        if isinstance(sample_val, str):
            if self.type == FieldType.STRING:
                return f'"{sample_val}"'
            return sample_val
        if isinstance(sample_val, bool):
            return "true" if sample_val else "false"
        return _go_fmt_value(sample_val)

    def set_sample(self, sample_val: Any) -> None:
        if self.type == FieldType.STRUCT:
            self.sample = f"{self.manifest_name}:"
        else:
            self.sample = (
                f"{self.manifest_name}: {self.get_sample_value(sample_val)}"
            )

    def set_default(self, sample_val: Any) -> None:
        self.default = self.get_sample_value(sample_val)
        if not self.markers:
            self.markers.extend(
                [
                    f"+kubebuilder:default={self.default}",
                    "+kubebuilder:validation:Optional",
                    f"(Default: {self.default})",
                ]
            )
        self.set_sample(sample_val)

    def set_comments_and_default(
        self, comments: Optional[list[str]], sample_val: Any, has_default: bool
    ) -> None:
        if has_default:
            self.set_default(sample_val)
        if comments is not None:
            self.comments.extend(comments)

    # ---- rendering -----------------------------------------------------

    def generate_api_spec(self, kind: str) -> str:
        out = [
            f"""
// {kind}Spec defines the desired state of {kind}.
type {kind}Spec struct {{
\t// INSERT ADDITIONAL SPEC FIELDS - desired state of cluster
\t// Important: Run "make" to regenerate code after modifying this file

"""
        ]

        for child in self.children:
            out.append(child._spec_field(kind))

        out.append("}\n\n")

        for child in self.children:
            if child.children:
                out.append(child._spec_struct(kind))

        return "".join(out)

    def _spec_field(self, kind: str) -> str:
        type_name = str(self.type)
        if self.type == FieldType.STRUCT:
            type_name = kind + self.struct_name

        out = []
        for m in self.markers:
            out.append(f"// {m}\n")
        for c in self.comments:
            out.append(f"// {c}\n")
        out.append(f"{self.name} {type_name} {self.tags}\n\n")
        return "".join(out)

    def _spec_struct(self, kind: str) -> str:
        if self.type != FieldType.STRUCT:
            return ""
        out = [f"type {kind}{self.struct_name} {self.type}{{\n"]
        for child in self.children:
            out.append(child._spec_field(kind))
        out.append("}\n\n")
        for child in self.children:
            out.append(child._spec_struct(kind))
        return "".join(out)

    def generate_sample_spec(self, required_only: bool) -> str:
        out: list[str] = []
        self._sample_spec(out, 0, required_only)
        return "".join(out)

    def _sample_spec(
        self, out: list[str], indent: int, required_only: bool
    ) -> None:
        out.append(f"{'  ' * indent}{self.sample}\n")
        for child in self.children:
            if child.needs_generate(required_only):
                child._sample_spec(out, indent + 1, required_only)

    def needs_generate(self, required_only: bool) -> bool:
        if not required_only:
            return True
        return self.has_required_field()

    def has_required_field(self) -> bool:
        if not self.children and self.default == "":
            return True
        return any(c.has_required_field() for c in self.children)
