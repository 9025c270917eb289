/*
Copyright 2021.

Licensed under the Apache License, Version 2.0 (the "License");
you may not use this file except in compliance with the License.
You may obtain a copy of the License at

    http://www.apache.org/licenses/LICENSE-2.0

Unless required by applicable law or agreed to in writing, software
distributed under the License is distributed on an "AS IS" BASIS,
WITHOUT WARRANTIES OR CONDITIONS OF ANY KIND, either express or implied.
See the License for the specific language governing permissions and
limitations under the License.
*/

package v1alpha1

import (
	"errors"

	"github.com/nukleros/operator-builder-tools/pkg/controller/workload"
	"github.com/nukleros/operator-builder-tools/pkg/status"
	metav1 "k8s.io/apimachinery/pkg/apis/meta/v1"
	"k8s.io/apimachinery/pkg/runtime/schema"
)

var ErrUnableToConvertEdgePlatform = errors.New("unable to convert to EdgePlatform")

// EDIT THIS FILE!  THIS IS SCAFFOLDING FOR YOU TO OWN!
// NOTE: json tags are required.  Any new fields you add must have json tags for the fields to be serialized.

// EdgePlatformSpec defines the desired state of EdgePlatform.
type EdgePlatformSpec struct {
	// INSERT ADDITIONAL SPEC FIELDS - desired state of cluster
	// Important: Run "make" to regenerate code after modifying this file

	// +kubebuilder:default="us-east-1"
	// +kubebuilder:validation:Optional
	// (Default: "us-east-1")
	Region string `json:"region,omitempty"`

	// +kubebuilder:default="standard"
	// +kubebuilder:validation:Optional
	// (Default: "standard")
	// platform service tier
	Tier string `json:"tier,omitempty"`

	// +kubebuilder:default="envoy:v1.22"
	// +kubebuilder:validation:Optional
	// (Default: "envoy:v1.22")
	GatewayImage string `json:"gatewayImage,omitempty"`
}

// EdgePlatformStatus defines the observed state of EdgePlatform.
type EdgePlatformStatus struct {
	// INSERT ADDITIONAL STATUS FIELD - define observed state of cluster
	// Important: Run "make" to regenerate code after modifying this file

	Created               bool                       `json:"created,omitempty"`
	DependenciesSatisfied bool                       `json:"dependenciesSatisfied,omitempty"`
	Conditions            []*status.PhaseCondition   `json:"conditions,omitempty"`
	Resources             []*status.ChildResource    `json:"resources,omitempty"`
}

// +kubebuilder:object:root=true
// +kubebuilder:subresource:status
// +kubebuilder:resource:scope=Cluster

// EdgePlatform is the Schema for the edgeplatforms API.
type EdgePlatform struct {
	metav1.TypeMeta   `json:",inline"`
	metav1.ObjectMeta `json:"metadata,omitempty"`
	Spec   EdgePlatformSpec   `json:"spec,omitempty"`
	Status EdgePlatformStatus `json:"status,omitempty"`
}

// +kubebuilder:object:root=true

// EdgePlatformList contains a list of EdgePlatform.
type EdgePlatformList struct {
	metav1.TypeMeta `json:",inline"`
	metav1.ListMeta `json:"metadata,omitempty"`
	Items           []EdgePlatform `json:"items"`
}

// interface methods

// GetReadyStatus returns the ready status for a component.
func (component *EdgePlatform) GetReadyStatus() bool {
	return component.Status.Created
}

// SetReadyStatus sets the ready status for a component.
func (component *EdgePlatform) SetReadyStatus(ready bool) {
	component.Status.Created = ready
}

// GetDependencyStatus returns the dependency status for a component.
func (component *EdgePlatform) GetDependencyStatus() bool {
	return component.Status.DependenciesSatisfied
}

// SetDependencyStatus sets the dependency status for a component.
func (component *EdgePlatform) SetDependencyStatus(dependencyStatus bool) {
	component.Status.DependenciesSatisfied = dependencyStatus
}

// GetPhaseConditions returns the phase conditions for a component.
func (component *EdgePlatform) GetPhaseConditions() []*status.PhaseCondition {
	return component.Status.Conditions
}

// SetPhaseCondition sets the phase conditions for a component.
func (component *EdgePlatform) SetPhaseCondition(condition *status.PhaseCondition) {
	for i, currentCondition := range component.GetPhaseConditions() {
		if currentCondition.Phase == condition.Phase {
			component.Status.Conditions[i] = condition

			return
		}
	}

	// phase not found, lets add it to the list.
	component.Status.Conditions = append(component.Status.Conditions, condition)
}

// GetResources returns the child resource status for a component.
func (component *EdgePlatform) GetChildResourceConditions() []*status.ChildResource {
	return component.Status.Resources
}

// SetResources sets the phase conditions for a component.
func (component *EdgePlatform) SetChildResourceCondition(resource *status.ChildResource) {
	for i, currentResource := range component.GetChildResourceConditions() {
		if currentResource.Group == resource.Group && currentResource.Version == resource.Version && currentResource.Kind == resource.Kind {
			if currentResource.Name == resource.Name && currentResource.Namespace == resource.Namespace {
				component.Status.Resources[i] = resource

				return
			}
		}
	}

	// phase not found, lets add it to the collection
	component.Status.Resources = append(component.Status.Resources, resource)
}

// GetDependencies returns the dependencies for a component.
func (*EdgePlatform) GetDependencies() []workload.Workload {
	return []workload.Workload{
	}
}

// GetComponentGVK returns a GVK object for the component.
func (*EdgePlatform) GetWorkloadGVK() schema.GroupVersionKind {
	return GroupVersion.WithKind("EdgePlatform")
}

func init() {
	SchemeBuilder.Register(&EdgePlatform{}, &EdgePlatformList{})
}
