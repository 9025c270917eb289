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

package edgeplatform

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	edgeplatformv1alpha1 "github.com/acme/edgeplatform/apis/edgeplatform/v1alpha1"
)

// +kubebuilder:rbac:groups=core,resources=configmaps,verbs=get;list;watch;create;update;patch;delete

const ConfigMapEdgeSystemEdgePlatformSettings = "edge-platform-settings"

// CreateConfigMapEdgeSystemEdgePlatformSettings creates the edge-platform-settings ConfigMap resource.
func CreateConfigMapEdgeSystemEdgePlatformSettings(
	parent *edgeplatformv1alpha1.EdgePlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "ConfigMap",
			"metadata": map[string]interface{}{
				"name": "edge-platform-settings",
				"namespace": "edge-system",
			},
			"data": map[string]interface{}{
				"region": parent.Spec.Region,
				"tier": parent.Spec.Tier,
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
