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

var ErrUnableToConvertEdgeApp = errors.New("unable to convert to EdgeApp")

// EDIT THIS FILE!  THIS IS SCAFFOLDING FOR YOU TO OWN!
// NOTE: json tags are required.  Any new fields you add must have json tags for the fields to be serialized.

// EdgeAppSpec defines the desired state of EdgeApp.
type EdgeAppSpec struct {
	// INSERT ADDITIONAL SPEC FIELDS - desired state of cluster
	// Important: Run "make" to regenerate code after modifying this file

	// +kubebuilder:default="aws"
	// +kubebuilder:validation:Optional
	// (Default: "aws")
	//  +kubebuilder:validation:Enum=aws;azure;gcp
	//  Cloud provider selection; semicolons in kubebuilder markers
	//  must survive the marker lexer.
	Cloud string `json:"cloud,omitempty"`

	// +kubebuilder:default=2
	// +kubebuilder:validation:Optional
	// (Default: 2)
	EdgeReplicas int `json:"edgeReplicas,omitempty"`

	// +kubebuilder:default=true
	// +kubebuilder:validation:Optional
	// (Default: true)
	UseFqdn bool `json:"useFqdn,omitempty"`
}

// EdgeAppStatus defines the observed state of EdgeApp.
type EdgeAppStatus struct {
	// INSERT ADDITIONAL STATUS FIELD - define observed state of cluster
	// Important: Run "make" to regenerate code after modifying this file

	Created               bool                       `json:"created,omitempty"`
	DependenciesSatisfied bool                       `json:"dependenciesSatisfied,omitempty"`
	Conditions            []*status.PhaseCondition   `json:"conditions,omitempty"`
	Resources             []*status.ChildResource    `json:"resources,omitempty"`
}

// +kubebuilder:object:root=true
// +kubebuilder:subresource:status

// EdgeApp is the Schema for the edgeapps API.
type EdgeApp struct {
	metav1.TypeMeta   `json:",inline"`
	metav1.ObjectMeta `json:"metadata,omitempty"`
	Spec   EdgeAppSpec   `json:"spec,omitempty"`
	Status EdgeAppStatus `json:"status,omitempty"`
}

// +kubebuilder:object:root=true

// EdgeAppList contains a list of EdgeApp.
type EdgeAppList struct {
	metav1.TypeMeta `json:",inline"`
	metav1.ListMeta `json:"metadata,omitempty"`
	Items           []EdgeApp `json:"items"`
}

// interface methods

// GetReadyStatus returns the ready status for a component.
func (component *EdgeApp) GetReadyStatus() bool {
	return component.Status.Created
}

// SetReadyStatus sets the ready status for a component.
func (component *EdgeApp) SetReadyStatus(ready bool) {
	component.Status.Created = ready
}

// GetDependencyStatus returns the dependency status for a component.
func (component *EdgeApp) GetDependencyStatus() bool {
	return component.Status.DependenciesSatisfied
}

// SetDependencyStatus sets the dependency status for a component.
func (component *EdgeApp) SetDependencyStatus(dependencyStatus bool) {
	component.Status.DependenciesSatisfied = dependencyStatus
}

// GetPhaseConditions returns the phase conditions for a component.
func (component *EdgeApp) GetPhaseConditions() []*status.PhaseCondition {
	return component.Status.Conditions
}

// SetPhaseCondition sets the phase conditions for a component.
func (component *EdgeApp) SetPhaseCondition(condition *status.PhaseCondition) {
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
func (component *EdgeApp) GetChildResourceConditions() []*status.ChildResource {
	return component.Status.Resources
}

// SetResources sets the phase conditions for a component.
func (component *EdgeApp) SetChildResourceCondition(resource *status.ChildResource) {
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
func (*EdgeApp) GetDependencies() []workload.Workload {
	return []workload.Workload{
	}
}

// GetComponentGVK returns a GVK object for the component.
func (*EdgeApp) GetWorkloadGVK() schema.GroupVersionKind {
	return GroupVersion.WithKind("EdgeApp")
}

func init() {
	SchemeBuilder.Register(&EdgeApp{}, &EdgeAppList{})
}
