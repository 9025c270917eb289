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

var ErrUnableToConvertBookStore = errors.New("unable to convert to BookStore")

// EDIT THIS FILE!  THIS IS SCAFFOLDING FOR YOU TO OWN!
// NOTE: json tags are required.  Any new fields you add must have json tags for the fields to be serialized.

// BookStoreSpec defines the desired state of BookStore.
type BookStoreSpec struct {
	// INSERT ADDITIONAL SPEC FIELDS - desired state of cluster
	// Important: Run "make" to regenerate code after modifying this file

	// +kubebuilder:validation:Optional
	Bookstore BookStoreSpecBookstore `json:"bookstore,omitempty"`

	// +kubebuilder:validation:Optional
	App BookStoreSpecApp `json:"app,omitempty"`

	// +kubebuilder:validation:Optional
	Service BookStoreSpecService `json:"service,omitempty"`
}

type BookStoreSpecBookstore struct{
	// +kubebuilder:validation:Optional
	Deeply BookStoreSpecBookstoreDeeply `json:"deeply,omitempty"`
}

type BookStoreSpecBookstoreDeeply struct{
	// +kubebuilder:validation:Optional
	Nested BookStoreSpecBookstoreDeeplyNested `json:"nested,omitempty"`
}

type BookStoreSpecBookstoreDeeplyNested struct{
	// +kubebuilder:validation:Optional
	Path BookStoreSpecBookstoreDeeplyNestedPath `json:"path,omitempty"`

	// +kubebuilder:default="nginx:1.17"
	// +kubebuilder:validation:Optional
	// (Default: "nginx:1.17")
	// Defines the book store image
	Image string `json:"image,omitempty"`
}

type BookStoreSpecBookstoreDeeplyNestedPath struct{
	// +kubebuilder:default=2
	// +kubebuilder:validation:Optional
	// (Default: 2)
	Replicas int `json:"replicas,omitempty"`

	// +kubebuilder:validation:Optional
	Required BookStoreSpecBookstoreDeeplyNestedPathRequired `json:"required,omitempty"`
}

type BookStoreSpecBookstoreDeeplyNestedPathRequired struct{
	Port int `json:"port,omitempty"`
}

type BookStoreSpecApp struct{
	// +kubebuilder:default="bookstore"
	// +kubebuilder:validation:Optional
	// (Default: "bookstore")
	Label string `json:"label,omitempty"`
}

type BookStoreSpecService struct{
	// +kubebuilder:default="bookstore"
	// +kubebuilder:validation:Optional
	// (Default: "bookstore")
	Name string `json:"name,omitempty"`

	TargetPort int `json:"targetPort,omitempty"`
}

// BookStoreStatus defines the observed state of BookStore.
type BookStoreStatus struct {
	// INSERT ADDITIONAL STATUS FIELD - define observed state of cluster
	// Important: Run "make" to regenerate code after modifying this file

	Created               bool                       `json:"created,omitempty"`
	DependenciesSatisfied bool                       `json:"dependenciesSatisfied,omitempty"`
	Conditions            []*status.PhaseCondition   `json:"conditions,omitempty"`
	Resources             []*status.ChildResource    `json:"resources,omitempty"`
}

// +kubebuilder:object:root=true
// +kubebuilder:subresource:status

// BookStore is the Schema for the bookstores API.
type BookStore struct {
	metav1.TypeMeta   `json:",inline"`
	metav1.ObjectMeta `json:"metadata,omitempty"`
	Spec   BookStoreSpec   `json:"spec,omitempty"`
	Status BookStoreStatus `json:"status,omitempty"`
}

// +kubebuilder:object:root=true

// BookStoreList contains a list of BookStore.
type BookStoreList struct {
	metav1.TypeMeta `json:",inline"`
	metav1.ListMeta `json:"metadata,omitempty"`
	Items           []BookStore `json:"items"`
}

// interface methods

// GetReadyStatus returns the ready status for a component.
func (component *BookStore) GetReadyStatus() bool {
	return component.Status.Created
}

// SetReadyStatus sets the ready status for a component.
func (component *BookStore) SetReadyStatus(ready bool) {
	component.Status.Created = ready
}

// GetDependencyStatus returns the dependency status for a component.
func (component *BookStore) GetDependencyStatus() bool {
	return component.Status.DependenciesSatisfied
}

// SetDependencyStatus sets the dependency status for a component.
func (component *BookStore) SetDependencyStatus(dependencyStatus bool) {
	component.Status.DependenciesSatisfied = dependencyStatus
}

// GetPhaseConditions returns the phase conditions for a component.
func (component *BookStore) GetPhaseConditions() []*status.PhaseCondition {
	return component.Status.Conditions
}

// SetPhaseCondition sets the phase conditions for a component.
func (component *BookStore) SetPhaseCondition(condition *status.PhaseCondition) {
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
func (component *BookStore) GetChildResourceConditions() []*status.ChildResource {
	return component.Status.Resources
}

// SetResources sets the phase conditions for a component.
func (component *BookStore) SetChildResourceCondition(resource *status.ChildResource) {
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
func (*BookStore) GetDependencies() []workload.Workload {
	return []workload.Workload{
	}
}

// GetComponentGVK returns a GVK object for the component.
func (*BookStore) GetWorkloadGVK() schema.GroupVersionKind {
	return GroupVersion.WithKind("BookStore")
}

func init() {
	SchemeBuilder.Register(&BookStore{}, &BookStoreList{})
}
