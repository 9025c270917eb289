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

package cloudplatform

import (
	"k8s.io/apimachinery/pkg/apis/meta/v1/unstructured"
	"sigs.k8s.io/controller-runtime/pkg/client"

	platformsv1alpha1 "github.com/acme/platform/apis/platforms/v1alpha1"
)

// +kubebuilder:rbac:groups=core,resources=configmaps,verbs=get;list;watch;create;update;patch;delete

const ConfigMapPlatformSystemPlatformConfig = "platform-config"

// CreateConfigMapPlatformSystemPlatformConfig creates the platform-config ConfigMap resource.
func CreateConfigMapPlatformSystemPlatformConfig(
	parent *platformsv1alpha1.CloudPlatform,
) ([]client.Object, error) {
	resourceObjs := []client.Object{}
	var resourceObj = &unstructured.Unstructured{
		Object: map[string]interface{}{
			"apiVersion": "v1",
			"kind": "ConfigMap",
			"metadata": map[string]interface{}{
				"name": "platform-config",
				"namespace": "platform-system",
			},
			"data": map[string]interface{}{
				"environment": parent.Spec.Environment,
				"telemetry": parent.Spec.Telemetry,
			},
		},
	}

	resourceObjs = append(resourceObjs, resourceObj)

	return resourceObjs, nil
}
